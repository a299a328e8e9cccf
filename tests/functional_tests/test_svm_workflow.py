"""Reference analog: tests/functional_tests/svm_workflow.py — a 3-step
ML pipeline where only training runs over SSH (exercises DepsPip
injection and a non-trivial pickled model return)."""

import pytest

ct = pytest.importorskip("covalent")
pytest.importorskip("sklearn")

pytestmark = pytest.mark.functional_tests


def test_svm_workflow():
    import numpy as np

    deps = ct.DepsPip(packages=["scikit-learn"])

    @ct.electron
    def make_data(n=200):
        rng = np.random.default_rng(0)
        x = rng.normal(size=(n, 4))
        y = (x[:, 0] + x[:, 1] > 0).astype(int)
        return x, y

    @ct.electron(executor="ssh", deps_pip=deps)
    def train_svm(data):
        from sklearn.svm import SVC

        x, y = data
        clf = SVC(kernel="linear").fit(x, y)
        return clf

    @ct.electron
    def score(clf, data):
        x, y = data
        return clf.score(x, y)

    @ct.lattice
    def workflow():
        data = make_data()
        clf = train_svm(data)
        return score(clf, data)

    dispatch_id = ct.dispatch(workflow)()
    result = ct.get_result(dispatch_id, wait=True)
    assert str(result.status) == "COMPLETED"
    assert result.result > 0.8
