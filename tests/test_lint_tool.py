"""tools/lint.py self-tests: each rule fires on a minimal offender and
stays quiet on clean code / noqa suppressions."""

import pathlib
import subprocess
import sys

REPO = pathlib.Path(__file__).resolve().parent.parent


def run_lint(tmp_path, source, name="sample.py"):
    f = tmp_path / name
    f.write_text(source)
    out = subprocess.run(
        [sys.executable, str(REPO / "tools" / "lint.py"), str(f)],
        capture_output=True, text=True,
    )
    return out.returncode, out.stdout


def test_unused_import_f401(tmp_path):
    rc, out = run_lint(tmp_path, "import os\nimport sys\n\nprint(sys.path)\n")
    assert rc == 1 and "F401" in out and "os" in out


def test_used_import_clean(tmp_path):
    rc, out = run_lint(tmp_path, "import os\n\nprint(os.sep)\n")
    assert rc == 0, out


def test_mutable_default_b006(tmp_path):
    rc, out = run_lint(tmp_path, "def f(x=[]):\n    return x\n")
    assert rc == 1 and "B006" in out


def test_bare_except_e722(tmp_path):
    rc, out = run_lint(
        tmp_path, "try:\n    pass\nexcept:\n    pass\n")
    assert rc == 1 and "E722" in out


def test_fstring_without_placeholder_f502(tmp_path):
    rc, out = run_lint(tmp_path, 'x = f"hello"\nprint(x)\n')
    assert rc == 1 and "F502" in out


def test_format_spec_not_flagged(tmp_path):
    # format specs are JoinedStr nodes; must not false-positive
    rc, out = run_lint(tmp_path, 'v = 1.5\nprint(f"{v:.2f}")\n')
    assert rc == 0, out


def test_noqa_suppresses_specific_code(tmp_path):
    rc, out = run_lint(
        tmp_path, "import os  # noqa: F401 (re-export)\n")
    assert rc == 0, out


def test_noqa_wrong_code_does_not_suppress(tmp_path):
    rc, out = run_lint(tmp_path, "import os  # noqa: E722\n")
    assert rc == 1 and "F401" in out


def test_syntax_error_e999(tmp_path):
    rc, out = run_lint(tmp_path, "def broken(:\n")
    assert rc == 1 and "E999" in out


def test_redefinition_f811(tmp_path):
    rc, out = run_lint(
        tmp_path, "from os.path import join\n\ndef join(a, b):\n    return a + b\n")
    assert rc == 1 and "F811" in out


def test_repo_is_lint_clean():
    out = subprocess.run(
        [sys.executable, str(REPO / "tools" / "lint.py")],
        capture_output=True, text=True, cwd=REPO,
    )
    assert out.returncode == 0, out.stdout
