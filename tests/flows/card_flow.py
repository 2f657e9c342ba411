from metaflow_amd import FlowSpec, card, current, step
from metaflow_amd.plugins.card_components import (
    Artifact,
    Image,
    Markdown,
    Table,
)

# 1x1 red PNG
_PNG = bytes.fromhex(
    "89504e470d0a1a0a0000000d49484452000000010000000108020000009077"
    "53de0000000c4944415408d763f8cfc0000000030001a7a24f9a0000000049"
    "454e44ae426082")


class CardFlow(FlowSpec):
    @card
    @step
    def start(self):
        self.metric = 0.125
        current.card.append("training summary", title="Notes")
        current.card.append("<b>custom html</b>")
        current.card.refresh()   # mid-task publish (progress card)
        current.card.extend([
            Markdown("## Results\n- loss **0.125**\n- `tokens/s` high"),
            Table([[1, 2], [3, 4]], headers=["a", "b"]),
            Image(_PNG, label="loss curve"),
            Artifact({"lr": 3e-4, "steps": 100}, name="config"),
        ])
        self.next(self.end)

    @card(id="report")
    @step
    def end(self):
        current.card.append(Markdown("# Final report"))


if __name__ == "__main__":
    CardFlow()
