"""Top-level `mfx` CLI: inspect flows/runs without a flow file.

Parity target: /root/reference/metaflow/cmd/ (`metaflow status`, etc.).

    python -m metaflow_amd status
    python -m metaflow_amd runs <FlowName>
    python -m metaflow_amd logs <Flow/run/step/task>
    python -m metaflow_amd card <Flow/run/step/task>
    python -m metaflow_amd gpus
"""

import click


@click.group()
def cli():
    pass


@cli.command(help="List flows in the local datastore.")
def status():
    from .client import Metaflow, namespace

    namespace(None)
    flows = list(Metaflow())
    if not flows:
        click.echo("No flows found (datastore root empty).")
        return
    for flow in flows:
        run = flow.latest_run
        click.echo("%-30s latest run: %s  (%s)"
                   % (flow.id, run.id if run else "-",
                      "ok" if run and run.successful else "…"))


@cli.command(help="List runs of a flow.")
@click.argument("flow_name")
@click.option("--limit", default=10)
def runs(flow_name, limit):
    from .client import Flow, namespace

    namespace(None)
    for i, run in enumerate(Flow(flow_name)):
        if i >= limit:
            break
        click.echo("%-22s %-10s tags=%s"
                   % (run.id,
                      "ok" if run.successful else "failed/running",
                      ",".join(t for t in run.tags
                               if not t.startswith("user:"))))


@cli.command(help="Show merged logs of a task pathspec.")
@click.argument("pathspec")
@click.option("--stderr", is_flag=True)
def logs(pathspec, stderr):
    from .client import Task, namespace

    namespace(None)
    task = Task(pathspec)
    click.echo(task.stderr if stderr else task.stdout)


@cli.command(help="Write a task's HTML card to a file.")
@click.argument("pathspec")
@click.option("--out", default="card.html")
def card(pathspec, out):
    from .client import Task, namespace
    from .plugins.card_decorator import get_card

    namespace(None)
    task = Task(pathspec)
    html = get_card(task._ds)
    if not html:
        raise click.ClickException("No card for %s" % pathspec)
    with open(out, "w") as f:
        f.write(html)
    click.echo("wrote %s" % out)


@cli.command(name="card-server",
             help="Serve task cards over HTTP for browsing.")
@click.option("--port", default=8324)
def card_server(port):
    import http.server

    from .client import Metaflow, Task, namespace
    from .plugins.card_decorator import get_card

    namespace(None)

    class Handler(http.server.BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def _send(self, body, ctype="text/html"):
            self.send_response(200)
            self.send_header("Content-Type", ctype)
            self.end_headers()
            self.wfile.write(body.encode())

        def do_GET(self):
            path = self.path.strip("/")
            if not path:
                rows = []
                for flow in Metaflow():
                    run = flow.latest_run
                    if not run:
                        continue
                    for step in run:
                        for task in step:
                            html = get_card(task._ds)
                            if html:
                                rows.append(
                                    '<li><a href="/%s">%s</a></li>'
                                    % (task.pathspec, task.pathspec))
                self._send("<h1>mfx cards</h1><ul>%s</ul>"
                           % "".join(rows))
                return
            try:
                task = Task(path)
                html = get_card(task._ds)
                self._send(html or "<p>no card</p>")
            except Exception as e:
                self._send("<p>error: %s</p>" % e)

    server = http.server.ThreadingHTTPServer(("127.0.0.1", port), Handler)
    click.echo("serving cards at http://127.0.0.1:%d" % port)
    server.serve_forever()


@cli.command(help="Register a flow file as a named local deployment "
                  "(content-addressed code snapshot).")
@click.argument("flow_file")
@click.option("--name", default="prod")
def deploy(flow_file, name):
    from .runner import Deployer

    df = Deployer(flow_file).local().create(name=name)
    click.echo("deployed %s/%s  code=%s" % (df.flow_name, df.name,
                                            df.code_package_key[:16]))


@cli.command(help="List deployments of a flow.")
@click.argument("flow_name")
def deployments(flow_name):
    from .runner import DeployedFlow

    rows = DeployedFlow.list_deployed(flow_name)
    if not rows:
        click.echo("no deployments for %s" % flow_name)
        return
    for df in rows:
        click.echo("%-16s created %s  code=%s"
                   % (df.name, df.created_at, df.code_package_key[:16]))


@cli.command(help="Run a deployed flow's code snapshot; blocks until "
                  "done and prints the run id.")
@click.argument("flow_name")
@click.option("--name", default="prod")
@click.option("--param", "params", multiple=True,
              help="NAME=VALUE flow parameter (repeatable)")
def trigger(flow_name, name, params):
    from .runner import DeployedFlow

    df = DeployedFlow.get(flow_name, name)
    kwargs = {}
    for p in params:
        k, _sep, v = p.partition("=")
        kwargs[k] = v
    tr = df.trigger(**kwargs)
    tr.wait()
    click.echo("run %s: %s" % (tr.run_id, tr.status))


@cli.command(help="Show visible GPUs (rocm-smi summary).")
def gpus():
    try:
        import torch

        if torch.cuda.is_available():
            for i in range(torch.cuda.device_count()):
                p = torch.cuda.get_device_properties(i)
                click.echo("%d: %s  %.0f GB" % (
                    i, p.name, p.total_memory / 1e9))
        else:
            click.echo("No GPUs visible.")
    except Exception as e:  # noqa: BLE001
        click.echo("torch unavailable: %s" % e)


def main():
    cli(standalone_mode=True)


if __name__ == "__main__":
    main()
