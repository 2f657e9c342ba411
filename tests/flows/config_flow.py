from metaflow_amd import Config, FlowMutator, FlowSpec, step
from metaflow_amd.plugins.retry_decorator import RetryDecorator


class ApplyConfig(FlowMutator):
    """Adds @retry to every step when the config asks for it."""

    def mutate(self, mf):
        cfg = mf.configs.get("cfg")
        if cfg and cfg.get("retries", 0):
            for s in mf.steps:
                if not any(d.name == "retry" for d in s.decorators):
                    s.add_decorator(RetryDecorator,
                                    times=cfg.get("retries"))


@ApplyConfig()
class ConfigFlow(FlowSpec):
    cfg = Config("cfg", default_value={"scale": 3, "retries": 1})

    @step
    def start(self):
        self.value = 10 * self.cfg.scale
        self.next(self.end)

    @step
    def end(self):
        assert self.value == 10 * self.cfg.scale
        n_retry = [d.name for d in
                   type(self).end.decorators].count("retry")
        self.retry_attached = n_retry == 1


if __name__ == "__main__":
    ConfigFlow()
