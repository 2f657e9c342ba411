"""Static DAG extraction from a FlowSpec subclass.

The flow's structure is derived *statically* by parsing the AST of each
``@step`` method and finding its ``self.next(...)`` call — the same strategy
the reference uses (see /root/reference/metaflow/graph.py:95-333 for the
behavior being matched) but implemented fresh: we walk the class source once,
classify each node, and annotate split-parent stacks with an explicit
DFS instead of the reference's in-place traversal.

Node types:
  start / linear / end           plain transitions
  split                          self.next(self.a, self.b, ...)
  split-switch                   self.next(self.a, self.b, condition='var')
  foreach                        self.next(self.a, foreach='var')
  split-parallel                 self.next(self.a, num_parallel=N)  (gang step)
  join                           step signature has an `inputs` argument
"""

import ast
import inspect
import textwrap

def _deindent(src):
    return textwrap.dedent(src)


class DAGNode(object):
    def __init__(self, func_ast, decorators, doc, source_file, lineno_offset):
        self.name = func_ast.name
        self.func_lineno = func_ast.lineno + lineno_offset
        self.source_file = source_file
        self.decorators = decorators
        self.doc = doc

        # assigned by _parse
        self.type = None
        self.out_funcs = []
        self.has_tail_next = False
        self.invalid_tail_next = False
        self.num_args = 0
        self.foreach_param = None
        self.condition = None
        self.num_parallel = 0
        self.parallel_step = False
        self.tail_next_lineno = 0

        # assigned by FlowGraph._traverse
        self.in_funcs = set()
        self.split_parents = []
        self.matching_join = None

        self._parse(func_ast)

    def _expr_str(self, expr):
        return "%s.%s" % (expr.value.id, expr.attr)

    def _parse(self, func_ast):
        self.num_args = len(func_ast.args.args)
        tail = func_ast.body[-1]

        if self.name == "end":
            self.type = "end"
            return

        # every non-end step must end in self.next(...)
        try:
            if not (
                isinstance(tail, ast.Expr)
                and isinstance(tail.value, ast.Call)
                and isinstance(tail.value.func, ast.Attribute)
                and tail.value.func.attr == "next"
                and isinstance(tail.value.func.value, ast.Name)
                and tail.value.func.value.id == "self"
            ):
                return
            self.has_tail_next = True
            self.invalid_tail_next = True
            self.tail_next_lineno = tail.lineno
            call = tail.value
            self.out_funcs = [e.attr for e in call.args]

            keywords = {k.arg: k.value for k in call.keywords}
            if len(keywords) == 1:
                if "foreach" in keywords:
                    kv = keywords["foreach"]
                    if isinstance(kv, ast.Constant) and isinstance(kv.value, str):
                        self.type = "foreach"
                        self.foreach_param = kv.value
                        if len(self.out_funcs) == 1:
                            self.invalid_tail_next = False
                elif "num_parallel" in keywords:
                    self.type = "split-parallel"
                    self.parallel_step = True
                    kv = keywords["num_parallel"]
                    if isinstance(kv, ast.Constant):
                        self.num_parallel = kv.value
                    else:
                        self.num_parallel = -1  # runtime-determined
                    if len(self.out_funcs) == 1:
                        self.invalid_tail_next = False
                elif "condition" in keywords:
                    kv = keywords["condition"]
                    if isinstance(kv, ast.Constant) and isinstance(kv.value, str):
                        self.type = "split-switch"
                        self.condition = kv.value
                        if len(self.out_funcs) >= 1:
                            self.invalid_tail_next = False
            elif len(keywords) == 0:
                if len(self.out_funcs) == 1:
                    self.type = "start" if self.name == "start" else "linear"
                    self.invalid_tail_next = False
                elif len(self.out_funcs) > 1:
                    self.type = "split"
                    self.invalid_tail_next = False
        except AttributeError:
            return

        # a join is any step taking an extra `inputs` arg; its incoming type
        # overrides linear classification
        if self.num_args > 1 and self.type in ("linear", "start"):
            self.type = "join"
        if self.name == "start" and self.type != "split-switch":
            # start keeps its derived type for splits/foreach; mark specially
            if self.type == "linear":
                self.type = "start"

    def __str__(self):
        return "*[{0.name} {0.type} (p:{1})]*".format(
            self, " ".join(self.split_parents)
        )


class StepVisitor(ast.NodeVisitor):
    def __init__(self, nodes, flow, source_file, lineno_offset):
        self.nodes = nodes
        self.flow = flow
        self.source_file = source_file
        self.lineno_offset = lineno_offset
        super().__init__()

    def visit_FunctionDef(self, node):
        func = getattr(self.flow, node.name, None)
        if func and getattr(func, "is_step", False):
            decos = [deco.name for deco in getattr(func, "decorators", [])]
            self.nodes[node.name] = DAGNode(
                node, decos, ast.get_docstring(node), self.source_file,
                self.lineno_offset,
            )


class FlowGraph(object):
    def __init__(self, flow):
        self.name = flow.__name__
        self.nodes = self._create_nodes(flow)
        self._postprocess()
        self._traverse_graph()

    def _create_nodes(self, flow):
        nodes = {}
        # walk the MRO so inherited steps are picked up (child overrides win)
        for cls in reversed(inspect.getmro(flow)):
            if cls is object:
                continue
            try:
                source_file = inspect.getsourcefile(cls)
                src, lineno = inspect.getsourcelines(cls)
            except (TypeError, OSError):
                continue
            module_ast = ast.parse(_deindent("".join(src)))
            cls_ast = module_ast.body[0]
            visitor = StepVisitor(nodes, flow, source_file, lineno - 1)
            for stmt in cls_ast.body:
                if isinstance(stmt, (ast.FunctionDef, ast.AsyncFunctionDef)):
                    visitor.visit(stmt)
        return nodes

    def _postprocess(self):
        # a join whose parent is a join stays a join; fix joins mislabeled
        # as linear when num_args>1
        for node in self.nodes.values():
            if node.type == "linear" and node.num_args > 1:
                node.type = "join"

    def _traverse_graph(self):
        def traverse(node, seen, split_parents):
            if node.type in ("split", "foreach", "split-parallel"):
                node.split_parents = split_parents
                split_parents = split_parents + [node.name]
            elif node.type == "split-switch":
                # a switch is not a fan-out: exactly one branch runs, so
                # it has no matching join and must NOT deepen the split
                # stack — this is also what makes recursive switches
                # (back-edges) terminate here
                node.split_parents = split_parents
            elif node.type == "join":
                # the matching split is the innermost unjoined one
                if split_parents:
                    node.split_parents = split_parents[:-1]
                    node.matching_join_of = split_parents[-1]
                    self.nodes[split_parents[-1]].matching_join = node.name
                    split_parents = split_parents[:-1]
                else:
                    node.split_parents = []
            else:
                node.split_parents = split_parents

            for n in node.out_funcs:
                child = self.nodes.get(n)
                if child is None:
                    continue
                child.in_funcs.add(node.name)
                key = (n, tuple(split_parents))
                if key not in seen:
                    seen.add(key)
                    traverse(child, seen, split_parents)

        if "start" in self.nodes:
            traverse(self.nodes["start"], set(), [])

    def __getitem__(self, name):
        return self.nodes[name]

    def __contains__(self, name):
        return name in self.nodes

    def __iter__(self):
        return iter(self.nodes.values())

    def sorted_nodes(self):
        """Topological-ish order: BFS from start."""
        order = []
        seen = set()
        frontier = ["start"] if "start" in self.nodes else []
        while frontier:
            nxt = []
            for name in frontier:
                if name in seen or name not in self.nodes:
                    continue
                seen.add(name)
                order.append(name)
                nxt.extend(self.nodes[name].out_funcs)
            frontier = nxt
        # orphans last
        for name in self.nodes:
            if name not in seen:
                order.append(name)
        return order

    def output_dot(self):
        lines = ["digraph %s {" % self.name]
        for node in self:
            lines.append('  "%s" [label="%s\\n%s"];' % (node.name, node.name,
                                                        node.type))
            for out in node.out_funcs:
                lines.append('  "%s" -> "%s";' % (node.name, out))
        lines.append("}")
        return "\n".join(lines)

    def to_dict(self):
        return {
            node.name: {
                "type": node.type,
                "in_funcs": sorted(node.in_funcs),
                "out_funcs": node.out_funcs,
                "split_parents": node.split_parents,
                "matching_join": node.matching_join,
                "foreach_param": node.foreach_param,
                "condition": node.condition,
                "num_parallel": node.num_parallel,
                "doc": node.doc,
                "decorators": node.decorators,
            }
            for node in self
        }

    def __str__(self):
        return "\n".join(
            "%s => %s [%s]" % (n.name, ", ".join(n.out_funcs), n.type)
            for n in self
        )
