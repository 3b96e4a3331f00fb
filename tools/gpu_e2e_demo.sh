#!/bin/bash
# Full CLI flow on one MI355X: corpus -> index -> pack -> train -> warmstart
# -> full-checkpoint conversion. Writes everything under gpurun_out/e2e.
set -e
OUT=gpurun_out/e2e
rm -rf $OUT && mkdir -p $OUT
python - <<PY
import json, random
random.seed(0)
words = ["lorem", "ipsum", "dolor", "sit", "amet", "consectetur"]
with open("$OUT/corpus.jsonl", "w") as f:
    for i in range(256):
        f.write(json.dumps({"text": " ".join(random.choices(words, k=64))}) + "\n")
PY
python -m modalities_amd data create_raw_index $OUT/corpus.jsonl
cat > $OUT/pack.yaml <<YAML
settings:
  src_path: $OUT/corpus.jsonl
  dst_path: $OUT/corpus.pbin
  eod_token: "<eod>"
tokenizer:
  component_key: tokenizer
  variant_key: char
  config: {}
YAML
python -m modalities_amd data pack_encoded_data $OUT/pack.yaml
python - <<PY
from pathlib import Path
text = Path("tests/configs/config_tiny_e2e.yaml").read_text()
text = text.replace("DATASET_PATH_PLACEHOLDER", "$OUT/corpus.pbin")
text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", "$OUT/ckpt")
text = text.replace("RESULTS_PATH_PLACEHOLDER", "$OUT/evaluation_results.jsonl")
# GPU shapes: head_dim 64 for the HIP attention kernel
text = text.replace("vocab_size: 256", "vocab_size: 260")
text = text.replace("n_embd: 64", "n_embd: 256")
Path("$OUT/train.yaml").write_text(text)
PY
python -m modalities_amd run --config_file_path $OUT/train.yaml --test_comm
EXP=$(ls $OUT/ckpt | head -1)
echo "=== warmstart from $EXP ==="
python -m modalities_amd warmstart --config_file_path $OUT/train.yaml \
    --last_checkpoint_info_file_path $OUT/ckpt/$EXP/last_checkpoint_info.json || true
CKPT=$(python -c "import json; print(json.load(open('$OUT/ckpt/$EXP/last_checkpoint_info.json'))['checkpoint_folder_path'])")
python -m modalities_amd convert_checkpoint_to_full \
    --checkpoint_folder_path "$CKPT" --config_file_path $OUT/train.yaml \
    --output_path $OUT/model_full.pt
echo "=== results ==="
tail -3 $OUT/evaluation_results.jsonl
ls -la $OUT/model_full.pt
