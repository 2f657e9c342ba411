"""Graph extraction + lint unit tests (CPU)."""

import pytest

from metaflow_amd import FlowSpec, step
from metaflow_amd.exceptions import LintWarn
from metaflow_amd.graph import FlowGraph
from metaflow_amd.lint import lint


class Linear(FlowSpec):
    @step
    def start(self):
        self.next(self.mid)

    @step
    def mid(self):
        self.next(self.end)

    @step
    def end(self):
        pass


class Branch(FlowSpec):
    @step
    def start(self):
        self.next(self.a, self.b)

    @step
    def a(self):
        self.next(self.join)

    @step
    def b(self):
        self.next(self.join)

    @step
    def join(self, inputs):
        self.next(self.end)

    @step
    def end(self):
        pass


class Foreach(FlowSpec):
    @step
    def start(self):
        self.items = [1, 2]
        self.next(self.work, foreach="items")

    @step
    def work(self):
        self.next(self.join)

    @step
    def join(self, inputs):
        self.next(self.end)

    @step
    def end(self):
        pass


class Gang(FlowSpec):
    @step
    def start(self):
        self.next(self.train, num_parallel=4)

    @step
    def train(self):
        self.next(self.join)

    @step
    def join(self, inputs):
        self.next(self.end)

    @step
    def end(self):
        pass


def test_linear_graph():
    g = FlowGraph(Linear)
    assert g["start"].type == "start"
    assert g["mid"].type == "linear"
    assert g["end"].type == "end"
    assert g["start"].out_funcs == ["mid"]
    assert g["mid"].in_funcs == {"start"}
    lint(g)


def test_branch_graph():
    g = FlowGraph(Branch)
    assert g["start"].type == "split"
    assert g["join"].type == "join"
    assert g["join"].in_funcs == {"a", "b"}
    assert g["a"].split_parents == ["start"]
    assert g["start"].matching_join == "join"
    lint(g)


def test_foreach_graph():
    g = FlowGraph(Foreach)
    assert g["start"].type == "foreach"
    assert g["start"].foreach_param == "items"
    assert g["work"].split_parents == ["start"]
    lint(g)


def test_parallel_graph():
    g = FlowGraph(Gang)
    assert g["start"].type == "split-parallel"
    assert g["start"].num_parallel == 4
    lint(g)


def test_lint_missing_end():
    class NoEnd(FlowSpec):
        @step
        def start(self):
            self.next(self.mid)

        @step
        def mid(self):
            self.next(self.mid2)

        @step
        def mid2(self):
            pass

    g = FlowGraph(NoEnd)
    with pytest.raises(LintWarn):
        lint(g)


def test_lint_missing_next():
    class NoNext(FlowSpec):
        @step
        def start(self):
            pass

        @step
        def end(self):
            pass

    g = FlowGraph(NoNext)
    with pytest.raises(LintWarn):
        lint(g)


def test_lint_unjoined_split():
    class Unjoined(FlowSpec):
        @step
        def start(self):
            self.next(self.a, self.b)

        @step
        def a(self):
            self.next(self.end)

        @step
        def b(self):
            self.next(self.end)

        @step
        def end(self):
            pass

    g = FlowGraph(Unjoined)
    with pytest.raises(LintWarn):
        lint(g)


def test_lint_orphan_step():
    class Orphan(FlowSpec):
        @step
        def start(self):
            self.next(self.end)

        @step
        def lonely(self):
            self.next(self.end)

        @step
        def end(self):
            pass

    g = FlowGraph(Orphan)
    with pytest.raises(LintWarn):
        lint(g)


def test_graph_dict_and_dot():
    g = FlowGraph(Branch)
    d = g.to_dict()
    assert d["start"]["type"] == "split"
    assert "digraph" in g.output_dot()
    assert g.sorted_nodes()[0] == "start"
