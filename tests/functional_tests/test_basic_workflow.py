"""End-to-end lattice dispatch (reference analog:
tests/functional_tests/basic_workflow_test.py): a mixed ssh+local
lattice must COMPLETE; an electron that raises must FAIL the lattice."""

import pytest

ct = pytest.importorskip("covalent")

pytestmark = pytest.mark.functional_tests


def test_basic_workflow_completes():
    @ct.electron(executor="ssh")
    def join_words(a, b):
        return ", ".join([a, b])

    @ct.electron
    def excitement(phrase):
        return f"{phrase}!"

    @ct.lattice
    def simple_workflow(a, b):
        phrase = join_words(a, b)
        return excitement(phrase)

    dispatch_id = ct.dispatch(simple_workflow)("Hello", "World")
    result = ct.get_result(dispatch_id, wait=True)
    assert str(result.status) == "COMPLETED"
    assert result.result == "Hello, World!"


def test_failing_electron_fails_lattice():
    @ct.electron(executor="ssh")
    def boom():
        raise RuntimeError("intentional failure")

    @ct.lattice
    def failing_workflow():
        return boom()

    dispatch_id = ct.dispatch(failing_workflow)()
    result = ct.get_result(dispatch_id, wait=True)
    assert str(result.status) == "FAILED"


def test_gpu_probe_workflow():
    """MI355X-specific: the electron reports the GPU it was pinned to."""

    @ct.electron(executor="ssh")
    def gpu_identity():
        import os

        return {
            "slot": os.environ.get("CSP_GPU_SLOT"),
            "visible": os.environ.get("HIP_VISIBLE_DEVICES"),
        }

    @ct.lattice
    def probe_workflow():
        return gpu_identity()

    dispatch_id = ct.dispatch(probe_workflow)()
    result = ct.get_result(dispatch_id, wait=True)
    assert str(result.status) == "COMPLETED"
    assert result.result["slot"] is not None
