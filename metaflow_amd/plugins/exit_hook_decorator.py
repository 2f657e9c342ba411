"""@exit_hook flow decorator: run user functions in a fresh subprocess when
the run succeeds or fails.

Parity target: /root/reference/metaflow/plugins/exit_hook/ (hooks invoked
by the runtime at run end, runtime.py:997-1044).
"""

from ..decorators import FlowDecorator, make_flow_decorator


class ExitHookDecorator(FlowDecorator):
    name = "exit_hook"
    defaults = {"on_success": None, "on_failure": None}

    def run_hooks(self, flow_file, success, run_id):
        import subprocess
        import sys

        names = self.attributes["on_success" if success else "on_failure"]
        if not names:
            return
        if isinstance(names, str):
            names = [names]
        for fn_name in names:
            code = (
                "import importlib.util, sys\n"
                "spec = importlib.util.spec_from_file_location('f', %r)\n"
                "m = importlib.util.module_from_spec(spec)\n"
                "sys.modules['f'] = m\n"
                "spec.loader.exec_module(m)\n"
                "getattr(m, %r)(%r)\n" % (flow_file, fn_name, run_id)
            )
            subprocess.run([sys.executable, "-c", code], timeout=300)


exit_hook = make_flow_decorator(ExitHookDecorator)
