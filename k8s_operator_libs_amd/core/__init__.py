"""Kubernetes client substrate.

The reference (a Go library) sits on k8s.io/client-go and controller-runtime
(SURVEY.md L0).  This package is the from-scratch AMD-native equivalent:

- :mod:`meta` — helpers over JSON-shaped (dict) Kubernetes objects, label /
  field selector matching, JSON-merge-patch.
- :mod:`errors` — typed API errors (NotFound, Conflict, AlreadyExists).
- :mod:`fakecluster` — an in-memory kube-apiserver (objects, resourceVersions,
  optimistic concurrency, watches, pod eviction): the envtest equivalent used
  by the test suites and benchmarks.
- :mod:`client` — the typed ``Client`` interface (controller-runtime
  ``client.Client`` analogue) with the in-memory implementation.
- :mod:`restclient` — httpx implementation against a real apiserver.
- :mod:`apiserver` — FastAPI app exposing a FakeCluster over HTTP for
  wire-level tests of the REST client.
- :mod:`events` — ``record.EventRecorder`` analogue.
"""

from .errors import ApiError, NotFoundError, ConflictError, AlreadyExistsError  # noqa: F401
from .client import Client, FakeClient  # noqa: F401
from .fakecluster import FakeCluster  # noqa: F401
from .events import EventRecorder, FakeRecorder  # noqa: F401
