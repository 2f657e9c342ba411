import os

from metaflow_amd import FlowSpec, step


class ResumeFlow(FlowSpec):
    @step
    def start(self):
        self.counter_dir = os.environ["RESUME_COUNTER_DIR"]
        self._bump("start")
        self.x = 5
        self.next(self.middle)

    @step
    def middle(self):
        self._bump("middle")
        if os.environ.get("RESUME_FAIL") == "1":
            raise RuntimeError("failing so resume has something to do")
        self.y = self.x * 2
        self.next(self.end)

    @step
    def end(self):
        self._bump("end")
        self.z = self.y + 1

    def _bump(self, name):
        path = os.path.join(self.counter_dir, name)
        n = 0
        if os.path.exists(path):
            n = int(open(path).read())
        with open(path, "w") as f:
            f.write(str(n + 1))


if __name__ == "__main__":
    ResumeFlow()
