from metaflow_amd import FlowSpec, card, current, step


class CardFlow(FlowSpec):
    @card
    @step
    def start(self):
        self.metric = 0.125
        current.card.append("training summary", title="Notes")
        current.card.append("<b>custom html</b>")
        self.next(self.end)

    @step
    def end(self):
        pass


if __name__ == "__main__":
    CardFlow()
