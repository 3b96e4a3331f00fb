"""Library-usage example: extend the framework with a CUSTOM component
without forking it (capability parity with the reference's
tutorials/library_usage: register a user-defined loss under a new
component_key/variant_key, reference it from YAML, and run training
programmatically through Main).

Run on CPU:  python examples/library_usage/main.py
"""

from pathlib import Path

import numpy as np
import torch
from pydantic import BaseModel

from modalities_amd.batch import InferenceResultBatch
from modalities_amd.dataloader.packed_data import write_pbin
from modalities_amd.loss_functions import Loss
from modalities_amd.main import Main


class ClippedCrossEntropyLossConfig(BaseModel):
    target_key: str
    prediction_key: str
    clip_value: float
    tag: str = "ClippedCrossEntropyLoss"


class ClippedCrossEntropyLoss(Loss):
    """A user-defined loss: plain CLM cross-entropy, clamped at clip_value.

    Any class following the component protocol (constructor kwargs = the
    pydantic config fields) can be registered; nothing in the framework
    needs to change.
    """

    def __init__(self, target_key: str, prediction_key: str, clip_value: float,
                 tag: str = "ClippedCrossEntropyLoss"):
        super().__init__(tag)
        self.target_key = target_key
        self.prediction_key = prediction_key
        self.clip_value = clip_value

    def forward(self, forward_batch) -> torch.Tensor:
        if isinstance(forward_batch, InferenceResultBatch):
            labels = forward_batch.get_targets(self.target_key)
            logits = forward_batch.get_predictions(self.prediction_key)
        else:  # pipeline-schedule call signature (logits, labels)
            logits, labels = forward_batch
        loss = torch.nn.functional.cross_entropy(
            logits.view(-1, logits.shape[-1]).float(), labels.reshape(-1),
            ignore_index=-100)
        return torch.clamp(loss, max=self.clip_value)


def run(workdir: Path) -> Path:
    """Prepare tiny synthetic data, register the custom loss, train.

    Returns the path of the results JSONL the run wrote.
    """
    workdir.mkdir(parents=True, exist_ok=True)

    # 1) synthetic byte-token dataset in the framework's .pbin format
    rng = np.random.default_rng(11)
    docs = [rng.integers(0, 256, size=200, dtype=np.uint8) for _ in range(8)]
    pbin = workdir / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)

    # 2) config: note loss_fn uses component_key=loss,
    #    variant_key=clipped_cross_entropy — a key that does NOT exist in the
    #    default registry; we add it below.
    template = Path(__file__).parent / "config_library_usage.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(workdir / "checkpoints"))
    text = text.replace("RESULTS_PATH_PLACEHOLDER",
                        str(workdir / "evaluation_results.jsonl"))
    cfg = workdir / "config.yaml"
    cfg.write_text(text)

    # 3) register + build + run
    main_obj = Main(cfg, experiment_id="library_usage_example")
    main_obj.add_custom_component(
        component_key="loss", variant_key="clipped_cross_entropy",
        custom_component=ClippedCrossEntropyLoss,
        custom_config=ClippedCrossEntropyLossConfig)
    components = main_obj.build_components()
    assert isinstance(components.loss_fn, ClippedCrossEntropyLoss)
    main_obj.run(components)
    return workdir / "evaluation_results.jsonl"


if __name__ == "__main__":
    out = run(Path("out/library_usage"))
    print(f"done; results at {out}")
