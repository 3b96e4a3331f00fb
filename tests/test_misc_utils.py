"""OptimizersList/SchedulerList, TimeRecorder, param counting, CLI smoke,
HF adapter."""

import time

import pytest
import torch

from modalities_amd.models.gpt2 import GPT2LLM, GPT2LLMConfig
from modalities_amd.optimizers.optimizer_list import OptimizersList, SchedulerList
from modalities_amd.utils.util import (TimeRecorder,
                                       get_local_number_of_trainable_parameters,
                                       get_total_number_of_trainable_parameters)


def tiny():
    return GPT2LLM(GPT2LLMConfig(vocab_size=64, n_layer=1, n_head_q=2,
                                 n_head_kv=2, n_embd=32, ffn_hidden=64,
                                 sequence_length=16))


def test_optimizers_list_steps_all():
    m1, m2 = tiny(), tiny()
    o1 = torch.optim.SGD(m1.parameters(), lr=0.1)
    o2 = torch.optim.SGD(m2.parameters(), lr=0.1)
    opts = OptimizersList([o1, o2])
    for m in (m1, m2):
        out = m({"input_ids": torch.randint(0, 64, (1, 8))})["logits"]
        out.float().sum().backward()
    before = m1.wte.weight.clone()
    opts.step()
    assert not torch.equal(before, m1.wte.weight)
    opts.zero_grad()
    sd = opts.state_dict()
    assert len(sd["optimizers"]) == 2
    opts.load_state_dict(sd)
    s1 = torch.optim.lr_scheduler.StepLR(o1, 1)
    s2 = torch.optim.lr_scheduler.StepLR(o2, 1)
    sl = SchedulerList([s1, s2])
    sl.step()
    assert len(sl.get_last_lr()) == 2
    sl.load_state_dict(sl.state_dict())


def test_time_recorder():
    tr = TimeRecorder()
    with tr:
        time.sleep(0.01)
    assert tr.delta_t >= 0.01
    with pytest.raises(RuntimeError):
        tr.stop()
    tr.reset()
    assert tr.delta_t == 0.0


def test_param_counting_sharded_vs_plain():
    from modalities_amd.parallel.fsdp import XGMIShardedModel
    torch.manual_seed(0)
    model = tiny()
    plain = get_local_number_of_trainable_parameters(model)
    sharded = XGMIShardedModel.from_transformer(model, torch.device("cpu"),
                                                param_dtype=torch.float32)
    total = get_total_number_of_trainable_parameters(sharded)
    # flat units are padded to multiples of 64 per unit
    assert plain <= total <= plain + 64 * len(sharded.units)


def test_cli_help_and_data_commands(tmp_path):
    from click.testing import CliRunner

    from modalities_amd.__main__ import main as cli_main
    runner = CliRunner()
    res = runner.invoke(cli_main, ["--help"])
    assert res.exit_code == 0
    for cmd in ("run", "warmstart", "generate_text", "data", "benchmark",
                "profile", "convert_pytorch_to_hf_checkpoint"):
        assert cmd in res.output
    # data create_raw_index through the CLI
    src = tmp_path / "c.jsonl"
    src.write_text('{"text": "hello"}\n{"text": "world"}\n')
    res = runner.invoke(cli_main, ["data", "create_raw_index", str(src)])
    assert res.exit_code == 0, res.output
    assert "indexed 2 lines" in res.output


def test_hf_adapter_generate():
    pytest.importorskip("transformers")
    from modalities_amd.models.hf_adapter import get_hf_model_adapter
    torch.manual_seed(0)
    model = tiny()
    hf = get_hf_model_adapter(model)
    ids = torch.randint(0, 64, (1, 5))
    out = hf(input_ids=ids)
    assert out.logits.shape == (1, 5, 64)
    gen = hf.generate(ids, max_new_tokens=3, do_sample=False)
    assert gen.shape[1] == 8
