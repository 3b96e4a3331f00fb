"""The library-usage example must actually run: custom component registered
via Main.add_custom_component, referenced from YAML, trained end-to-end on
CPU (mirrors the reference's tutorials/library_usage)."""

import json
import sys
from pathlib import Path

EXAMPLE_DIR = Path(__file__).parent.parent / "examples" / "library_usage"


def test_library_usage_example_runs(tmp_path):
    sys.path.insert(0, str(EXAMPLE_DIR))
    try:
        import importlib
        mod = importlib.import_module("main")
        if not hasattr(mod, "ClippedCrossEntropyLoss"):  # name collision guard
            mod = importlib.reload(mod)
        results = mod.run(tmp_path / "work")
    finally:
        sys.path.remove(str(EXAMPLE_DIR))

    assert results.exists()
    records = [json.loads(ln) for ln in results.read_text().splitlines()]
    train = [r for r in records if r.get("dataloader_tag") == "train"]
    assert train, records
    # the custom loss clamps at 20.0, and the tag flows through to results
    for r in train:
        for key, value in r["losses"].items():
            v = value["value"] if isinstance(value, dict) else value
            assert v <= 20.0 + 1e-6, (key, value)
