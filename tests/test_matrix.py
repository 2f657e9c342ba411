"""Matrix harness: DAG shapes x behaviors x dual checkers."""

import pytest

from .matrix.harness import (
    GRAPHS,
    ArtifactFlow,
    CurrentInfoFlow,
    ForeachContextFlow,
    NumpyArtifactFlow,
    ResumeFlow,
    StepCounterFlow,
    run_matrix_case,
)

BEHAVIORS = {
    "artifact": ArtifactFlow,
    "counter": StepCounterFlow,
    "current": CurrentInfoFlow,
    "foreach_ctx": ForeachContextFlow,
    "resume": ResumeFlow,
    "numpy": NumpyArtifactFlow,
}

CASES = [(g, b) for g in GRAPHS for b in BEHAVIORS]


@pytest.mark.parametrize("graph_name,behavior", CASES,
                         ids=["%s-%s" % c for c in CASES])
def test_matrix(graph_name, behavior, tmp_path, tmp_datastore):
    test = BEHAVIORS[behavior]()
    run_matrix_case(graph_name, test, str(tmp_path), tmp_datastore)


API_CASES = [("linear", "artifact"), ("foreach", "counter"),
             ("nested_branch", "current")]


@pytest.mark.parametrize("graph_name,behavior", API_CASES,
                         ids=["api-%s-%s" % c for c in API_CASES])
def test_matrix_runner_api(graph_name, behavior, tmp_path, tmp_datastore):
    """Same generated flows driven through the Runner API executor
    (reference tier 1 runs every case under both executors)."""
    test = BEHAVIORS[behavior]()
    run_matrix_case(graph_name, test, str(tmp_path), tmp_datastore,
                    executor="api")
