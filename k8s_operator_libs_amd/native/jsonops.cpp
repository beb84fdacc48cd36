// jsonops.cpp — native accelerators for JSON-shaped Kubernetes objects.
//
// The in-memory apiserver's snapshot isolation deep-copies every object on
// get/list/watch, which profiling shows is the hottest primitive of the
// reconcile path (profiles/r01_bench_kernel_stats.md companion CPU profile).
// This CPython extension implements the copy natively for acyclic JSON trees
// (dict / list / immutable scalars) — no memo table, no reduce protocol.
//
// Native-runtime counterpart of the reference's compiled Go runtime: the
// control-plane hot loop runs on compiled code, Python stays the
// orchestration layer.

#define PY_SSIZE_T_CLEAN
#include <Python.h>

static PyObject* jsonops_deep_copy(PyObject* obj);

static PyObject* copy_dict(PyObject* src) {
  PyObject* dst = PyDict_New();
  if (!dst) return nullptr;
  PyObject *key, *value;
  Py_ssize_t pos = 0;
  while (PyDict_Next(src, &pos, &key, &value)) {
    PyObject* copied = jsonops_deep_copy(value);
    if (!copied) {
      Py_DECREF(dst);
      return nullptr;
    }
    // keys in JSON trees are strings (immutable): share them
    if (PyDict_SetItem(dst, key, copied) < 0) {
      Py_DECREF(copied);
      Py_DECREF(dst);
      return nullptr;
    }
    Py_DECREF(copied);
  }
  return dst;
}

static PyObject* copy_list(PyObject* src) {
  Py_ssize_t n = PyList_GET_SIZE(src);
  PyObject* dst = PyList_New(n);
  if (!dst) return nullptr;
  for (Py_ssize_t i = 0; i < n; ++i) {
    PyObject* copied = jsonops_deep_copy(PyList_GET_ITEM(src, i));
    if (!copied) {
      Py_DECREF(dst);
      return nullptr;
    }
    PyList_SET_ITEM(dst, i, copied);  // steals reference
  }
  return dst;
}

static PyObject* jsonops_deep_copy(PyObject* obj) {
  if (PyDict_CheckExact(obj)) return copy_dict(obj);
  if (PyList_CheckExact(obj)) return copy_list(obj);
  // scalars (str/int/float/bool/None) and anything exotic: share
  Py_INCREF(obj);
  return obj;
}

static PyObject* py_deep_copy(PyObject* /*self*/, PyObject* obj) {
  return jsonops_deep_copy(obj);
}

static PyMethodDef jsonops_methods[] = {
    {"deep_copy", py_deep_copy, METH_O,
     "Deep copy of a JSON-shaped tree (dict/list/scalars)."},
    {nullptr, nullptr, 0, nullptr},
};

static struct PyModuleDef jsonops_module = {
    PyModuleDef_HEAD_INIT, "_jsonops",
    "Native accelerators for JSON-shaped Kubernetes objects", -1,
    jsonops_methods,
};

PyMODINIT_FUNC PyInit__jsonops(void) { return PyModule_Create(&jsonops_module); }
