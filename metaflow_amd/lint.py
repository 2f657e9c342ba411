"""Static flow-graph validity checks, run before any execution.

Behavioral parity target: the ~17 checks in /root/reference/metaflow/lint.py
(lint.py:48-460). Implemented as a simple ordered list of check functions over
our FlowGraph.
"""

from .exceptions import LintWarn

RESERVED_STEP_NAMES = {"foreach", "joins", "input", "index", "next"}


class FlowLinter(object):
    def __init__(self):
        self._checks = []

    def check(self, f):
        self._checks.append(f)
        return f

    def run_checks(self, graph):
        for check in self._checks:
            check(graph)


linter = FlowLinter()


@linter.check
def check_reserved_words(graph):
    for node in graph:
        if node.name in RESERVED_STEP_NAMES:
            raise LintWarn(
                "Step name *%s* is a reserved word." % node.name,
                node.func_lineno,
            )


@linter.check
def check_basic_steps(graph):
    for name in ("start", "end"):
        if name not in graph:
            raise LintWarn("Flow must include a step named *%s*." % name)


@linter.check
def check_that_end_is_end(graph):
    node = graph["end"]
    if node.has_tail_next or node.out_funcs:
        raise LintWarn("The *end* step must not call self.next().",
                       node.func_lineno)
    if node.num_args > 1:
        raise LintWarn("The *end* step cannot be a join.", node.func_lineno)


@linter.check
def check_step_names(graph):
    for node in graph:
        if node.name.startswith("_"):
            raise LintWarn(
                "Step name *%s* is invalid: names must not start with an "
                "underscore." % node.name, node.func_lineno)


@linter.check
def check_num_args(graph):
    for node in graph:
        if node.num_args > 2:
            raise LintWarn(
                "Step *%s* takes too many arguments: steps accept only "
                "'self' (and 'inputs' for joins)." % node.name,
                node.func_lineno)
        if node.num_args == 2 and node.type != "join":
            raise LintWarn(
                "Step *%s* accepts an extra argument but is not a join: "
                "only joins take 'inputs'." % node.name, node.func_lineno)
        if node.num_args < 1:
            raise LintWarn("Step *%s* must take 'self' as its first argument."
                           % node.name, node.func_lineno)


@linter.check
def check_static_transitions(graph):
    for node in graph:
        if node.type != "end" and not node.has_tail_next:
            raise LintWarn(
                "Step *%s* must end with a self.next() transition."
                % node.name, node.func_lineno)


@linter.check
def check_valid_transitions(graph):
    for node in graph:
        if node.type != "end" and node.has_tail_next and \
                node.invalid_tail_next:
            raise LintWarn(
                "Step *%s* has an invalid self.next() transition. Valid "
                "forms: self.next(self.a), self.next(self.a, self.b), "
                "self.next(self.a, foreach='var'), "
                "self.next(self.a, num_parallel=N), "
                "self.next(self.a, self.b, condition='var')."
                % node.name, node.tail_next_lineno)


@linter.check
def check_unknown_transitions(graph):
    for node in graph:
        unknown = [n for n in node.out_funcs if n not in graph]
        if unknown:
            raise LintWarn(
                "Step *%s* transitions to unknown step(s): %s."
                % (node.name, ", ".join(unknown)), node.tail_next_lineno)


@linter.check
def check_for_orphans(graph):
    orphans = [
        n.name for n in graph
        if n.name != "start" and not n.in_funcs
    ]
    if orphans:
        raise LintWarn(
            "Step(s) %s are not reachable from *start*."
            % ", ".join(orphans))


@linter.check
def check_for_acyclicity(graph):
    # DFS cycle detection; recursive switch steps (a switch that can target
    # an ancestor) are the only allowed back-edges
    def dfs(name, path):
        node = graph[name]
        for out in node.out_funcs:
            if out in path:
                src = graph[name]
                if src.type == "split-switch":
                    continue  # recursive switch loop is allowed
                raise LintWarn(
                    "Cycle detected: step *%s* transitions back to *%s*."
                    % (name, out), node.tail_next_lineno)
            if out in graph:
                dfs(out, path + [out])

    if "start" in graph:
        dfs("start", ["start"])


@linter.check
def check_split_join_balance(graph):
    # LIFO split/join matching: every split must have a matching join,
    # every join must match a split
    def traverse(name, stack, seen):
        node = graph[name]
        if node.type in ("split", "split-switch", "foreach",
                         "split-parallel"):
            if node.type == "split-switch":
                # switch branches may rejoin without a join step (they
                # converge); don't force a join for switches
                new_stack = stack
            else:
                new_stack = stack + [node.name]
        elif node.type == "join":
            if not stack:
                raise LintWarn(
                    "Step *%s* is a join but there is no split to join."
                    % node.name, node.func_lineno)
            new_stack = stack[:-1]
        elif node.type == "end":
            if stack:
                raise LintWarn(
                    "Step *end* reached with unjoined split(s): %s. Every "
                    "split/foreach must be joined before *end*."
                    % ", ".join(stack))
            return
        else:
            new_stack = stack
        for out in node.out_funcs:
            key = (out, tuple(new_stack))
            if key in seen:
                continue
            seen.add(key)
            if out in graph and graph[out].type != "split-switch" or \
                    out in graph:
                traverse(out, new_stack, seen)

    if "start" in graph:
        # skip balance check entirely for flows containing recursive
        # switches (loops make the stack analysis ambiguous)
        has_switch_cycle = any(n.type == "split-switch" for n in graph)
        if not has_switch_cycle:
            traverse("start", [], set())


@linter.check
def check_empty_foreaches(graph):
    for node in graph:
        if node.type in ("foreach", "split-parallel"):
            joins = [n for n in node.out_funcs if graph[n].type == "join"]
            if joins:
                raise LintWarn(
                    "Step *%s* is a foreach split followed directly by a "
                    "join. Add at least one step inside the foreach."
                    % node.name, node.func_lineno)


@linter.check
def check_parallel_step_after_next(graph):
    # the target of num_parallel must not itself be a join
    for node in graph:
        if node.type == "split-parallel":
            for out in node.out_funcs:
                if graph[out].type == "join":
                    raise LintWarn(
                        "The target of a num_parallel transition cannot be "
                        "a join (step *%s*)." % out, node.func_lineno)


@linter.check
def check_join_inputs(graph):
    for node in graph:
        if node.type == "join":
            for in_name in node.in_funcs:
                src = graph[in_name]
                if src.type in ("split", "split-switch"):
                    raise LintWarn(
                        "Step *%s* joins *%s* directly after the split; "
                        "splits need at least one step per branch before "
                        "the join." % (node.name, in_name), node.func_lineno)


def lint(graph):
    linter.run_checks(graph)
