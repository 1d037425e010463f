"""Custom-reward PPO: replace the paired-RW reward model with an
EXTERNAL HuggingFace sequence classifier (reference counterpart:
examples/customized_exp/ppo_sentiment.py — a user-code extension that
registers its own ModelInterface for the rew_inf MFC).

Shows the extensibility contract: (1) write a ModelInterface whose
`inference` returns a SequenceSample with a per-sequence "rewards" key,
(2) register it under a name, (3) swap it into the built experiment's
interface table.  The rew role's weights are unused — the interface
carries its own scorer — so the reward ModelConfig can stay tiny.

Run (needs an HF classifier checkpoint, e.g. a distilbert sentiment
model, plus prompt data):

    SCORER=/path/to/hf-classifier python examples/customized_exp/ppo_sentiment.py \
        actor=... (same overrides as `quickstart ppo`)
"""
import dataclasses
import os
import sys
from typing import Optional

import torch

from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import Model, ModelInterface, register_interface


@dataclasses.dataclass
class SentimentScoringInterface(ModelInterface):
    """Scores full sequences with an external HF classifier; the
    positive-class logit becomes the PPO reward."""

    scorer_path: str = ""

    def __post_init__(self):
        import transformers

        path = self.scorer_path or os.environ["SCORER"]
        self.model = (
            transformers.AutoModelForSequenceClassification.from_pretrained(path)
        )
        if torch.cuda.is_available():
            self.model = self.model.cuda()
        self.model.eval()

    @torch.no_grad()
    def inference(self, model: Model, data: SequenceSample,
                  n_mbs: Optional[int] = None) -> SequenceSample:
        dev = next(self.model.parameters()).device
        ids = data.data["packed_input_ids"].to(dev)
        lens = [sum(x) for x in data.seqlens["packed_input_ids"]]
        mx = max(lens)
        # unpack to a right-padded [bs, mx] batch for the HF classifier
        bs = data.bs
        batch = torch.zeros(bs, mx, dtype=torch.long, device=dev)
        mask = torch.zeros(bs, mx, dtype=torch.long, device=dev)
        off = 0
        for i, l in enumerate(lens):
            batch[i, :l] = ids[off:off + l]
            mask[i, :l] = 1
            off += l
        logits = self.model(input_ids=batch, attention_mask=mask).logits
        # positive-class logit (last class) as the scalar reward
        scores = logits[:, -1].float().cpu()
        return SequenceSample(
            keys=("rewards",),
            ids=list(data.ids),
            seqlens={"rewards": [[1]] * bs},
            data={"rewards": scores},
        )


register_interface("sentiment_rw", SentimentScoringInterface)


def main(argv=None):
    from realhf_amd.apps.quickstart import parse_cli
    from realhf_amd.runtime.trainer import Trainer

    _, cfg = parse_cli(["ppo"] + list(argv or sys.argv[1:]))
    t = Trainer(cfg)
    # swap the reward interface for the custom scorer (the rew role's
    # own weights are unused by this interface)
    t.built.interfaces["rew_inf"] = SentimentScoringInterface(
        scorer_path=os.environ.get("SCORER", ""))
    t.run()


if __name__ == "__main__":
    main()
