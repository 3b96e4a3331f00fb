"""Config-driven TP: the YAML component path (device_mesh ->
tensor_parallelized_model -> sharded wrap -> Main.run) at world 2 on gloo —
the reference's YAML-driven parallelism flow (reference:
config_files/training/config_lorem_ipsum_long_fsdp2_tp.yaml family)."""

import json
from pathlib import Path

import numpy as np

from tests.utils_dist import run_distributed


def _rank_main(rank, world, cfg_path):
    from modalities_amd.main import Main
    main_obj = Main(Path(cfg_path), experiment_id="tp2_cfg")
    components = main_obj.build_components()
    main_obj.run(components)
    # read back rank-0's results JSONL
    results = Path(cfg_path).parent / "evaluation_results.jsonl"
    if rank == 0:
        records = [json.loads(ln) for ln in results.read_text().splitlines()]
        train = [r for r in records if r.get("dataloader_tag") == "train"]
        assert train, records
        return [next(iter(r["losses"].values())) for r in train]
    return None


def test_config_driven_tp2_trains(tmp_path):
    rng = np.random.default_rng(7)
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [rng.integers(0, 256, size=200, dtype=np.uint8) for _ in range(8)]
    pbin = tmp_path / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)

    template = Path(__file__).parent / "configs" / "config_tiny_e2e_tp2.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(tmp_path / "ckpt"))
    text = text.replace("RESULTS_PATH_PLACEHOLDER",
                        str(tmp_path / "evaluation_results.jsonl"))
    cfg = tmp_path / "config.yaml"
    cfg.write_text(text)

    results = run_distributed(_rank_main, world_size=2, port=29467,
                              args=(str(cfg),), timeout_s=420)
    losses = results[0]
    assert losses and all(np.isfinite(v) for v in losses), losses
    # checkpoints written
    assert (tmp_path / "ckpt" / "tp2_cfg" / "last_checkpoint_info.json").exists()


def _rank_main_pp(rank, world, cfg_path):
    from modalities_amd.main import Main
    main_obj = Main(Path(cfg_path), experiment_id="pp2_cfg")
    components = main_obj.build_components()
    assert components.pp_schedule is not None
    main_obj.run(components)
    return "ok"


def test_config_driven_pp2_trains(tmp_path):
    """YAML-driven PP: device_mesh(pp=2) -> staged schedule ->
    pipelined_model selector -> Main.run at world 2 on gloo."""
    rng = np.random.default_rng(9)
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [rng.integers(0, 256, size=200, dtype=np.uint8) for _ in range(8)]
    pbin = tmp_path / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)

    template = Path(__file__).parent / "configs" / "config_tiny_e2e_pp2.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(tmp_path / "ckpt"))
    text = text.replace("RESULTS_PATH_PLACEHOLDER",
                        str(tmp_path / "evaluation_results.jsonl"))
    cfg = tmp_path / "config.yaml"
    cfg.write_text(text)

    results = run_distributed(_rank_main_pp, world_size=2, port=29473,
                              args=(str(cfg),), timeout_s=420)
    assert results == {0: "ok", 1: "ok"}
    # each PP partition saved its own meta + shard namespace
    ckpt_root = tmp_path / "ckpt" / "pp2_cfg"
    folders = [p for p in ckpt_root.iterdir() if p.is_dir()]
    assert folders
    names = {f.name for f in folders[0].iterdir()}
    assert "meta_pp0.json" in names and "meta_pp1.json" in names, names


def _rank_main_cp(rank, world, cfg_path):
    from modalities_amd.main import Main
    main_obj = Main(Path(cfg_path), experiment_id="cp2_cfg")
    components = main_obj.build_components()
    main_obj.run(components)
    return "ok"


def test_config_driven_cp2_trains(tmp_path):
    """YAML-driven CP: device_mesh(cp=2) + context_parallelized_model;
    the trainer slices targets per CP rank and rescales the loss."""
    rng = np.random.default_rng(13)
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [rng.integers(0, 256, size=200, dtype=np.uint8) for _ in range(8)]
    pbin = tmp_path / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)

    template = Path(__file__).parent / "configs" / "config_tiny_e2e_cp2.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(tmp_path / "ckpt"))
    text = text.replace("RESULTS_PATH_PLACEHOLDER",
                        str(tmp_path / "evaluation_results.jsonl"))
    cfg = tmp_path / "config.yaml"
    cfg.write_text(text)

    results = run_distributed(_rank_main_cp, world_size=2, port=29479,
                              args=(str(cfg),), timeout_s=420)
    assert results == {0: "ok", 1: "ok"}
    records = [json.loads(ln)
               for ln in (tmp_path / "evaluation_results.jsonl").read_text().splitlines()]
    assert any(r.get("dataloader_tag") == "train" for r in records)


def _rank_main_tp2dp2(rank, world, cfg_path):
    from modalities_amd.main import Main
    main_obj = Main(Path(cfg_path), experiment_id="tp2dp2_cfg")
    components = main_obj.build_components()
    main_obj.run(components)
    return "ok"


def test_config_driven_tp2_dp2_world4(tmp_path):
    """World-4 composition from YAML: TP2 x DP2 mesh degrees + TP stage +
    sharded wrap + samplers keyed off the dp dims of the mesh."""
    rng = np.random.default_rng(21)
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [rng.integers(0, 256, size=400, dtype=np.uint8) for _ in range(10)]
    pbin = tmp_path / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)

    template = Path(__file__).parent / "configs" / "config_tiny_e2e_tp2dp2.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(tmp_path / "ckpt"))
    text = text.replace("RESULTS_PATH_PLACEHOLDER",
                        str(tmp_path / "evaluation_results.jsonl"))
    cfg = tmp_path / "config.yaml"
    cfg.write_text(text)

    results = run_distributed(_rank_main_tp2dp2, world_size=4, port=29487,
                              args=(str(cfg),), timeout_s=420)
    assert all(v == "ok" for v in results.values()) and len(results) == 4
    assert (tmp_path / "ckpt" / "tp2dp2_cfg" / "last_checkpoint_info.json").exists()


def _rank_sampler_identity(rank, world):
    """TP peers (same dp coords) must enumerate IDENTICAL sample indices."""
    import torch.distributed as dist

    from modalities_amd.parallel.mesh import get_device_mesh
    from modalities_amd.registry.components import get_resumable_sampler

    mesh = get_device_mesh(world, rank, tensor_parallel_degree=2)
    sampler = get_resumable_sampler(list(range(64)), shuffle=True, seed=3,
                                    device_mesh=mesh)
    idx = list(sampler)
    gathered = [None] * world
    dist.all_gather_object(gathered, (mesh.dp_rank, idx))
    by_dp = {}
    for dp_rank, indices in gathered:
        if dp_rank in by_dp:
            assert by_dp[dp_rank] == indices, "TP peers saw different data"
        else:
            by_dp[dp_rank] = indices
    # dp groups partition the data (no overlap across dp ranks)
    all_sets = [set(v) for v in by_dp.values()]
    assert not set.intersection(*all_sets) if len(all_sets) > 1 else True
    return "ok"


def test_mesh_sampler_tp_peers_share_data():
    results = run_distributed(_rank_sampler_identity, world_size=4, port=29491)
    assert all(v == "ok" for v in results.values())


def _rank_warmstart_pp(rank, world, cfg_path, ws_cfg_path):
    import yaml

    from modalities_amd.main import Main

    # run A: train 8 steps with PP2, checkpointing every 4
    main_a = Main(Path(cfg_path), experiment_id="ppA")
    main_a.run(main_a.build_components())

    # find the step-4 checkpoint
    root = Path(cfg_path).parent / "ckpt" / "ppA"
    step4 = [p for p in root.iterdir() if "seen_steps_4" in p.name][0]

    # run B: warmstart from it through the config path
    def ws_resolver(key):
        return {"checkpoint_folder_path": str(step4)}[key]

    main_b = Main(Path(ws_cfg_path), experiment_id="ppB",
                  additional_resolver_funs={"warmstart_env": ws_resolver})
    main_b.run(main_b.build_components())
    return "ok"


def test_config_driven_pp2_warmstart(tmp_path):
    """Config-path warmstart under PP2: each partition loads its own
    meta/shard namespace via the mesh-aware app_state factory."""
    import yaml

    rng = np.random.default_rng(31)
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [rng.integers(0, 256, size=200, dtype=np.uint8) for _ in range(8)]
    pbin = tmp_path / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)

    template = Path(__file__).parent / "configs" / "config_tiny_e2e_pp2.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(tmp_path / "ckpt"))
    text = text.replace("RESULTS_PATH_PLACEHOLDER",
                        str(tmp_path / "a_results.jsonl"))
    cfg = tmp_path / "config.yaml"
    cfg.write_text(text)

    # warmstart config: swap app_state for the mesh-aware sharded_warmstart
    cfg_dict = yaml.safe_load(text.replace(str(tmp_path / "a_results.jsonl"),
                                           str(tmp_path / "b_results.jsonl")))
    cfg_dict["app_state"] = {
        "component_key": "app_state", "variant_key": "sharded_warmstart",
        "config": {
            "model": {"instance_key": "wrapped_model", "pass_type": "BY_REFERENCE"},
            "optimizer": {"instance_key": "optimizer", "pass_type": "BY_REFERENCE"},
            "lr_scheduler": {"instance_key": "scheduler", "pass_type": "BY_REFERENCE"},
            "checkpoint_folder_path": "${warmstart_env:checkpoint_folder_path}",
            "device_mesh": {"instance_key": "device_mesh",
                            "pass_type": "BY_REFERENCE"},
        },
    }
    ws_cfg = tmp_path / "ws.yaml"
    ws_cfg.write_text(yaml.safe_dump(cfg_dict, sort_keys=False))

    results = run_distributed(_rank_warmstart_pp, world_size=2, port=29497,
                              args=(str(cfg), str(ws_cfg)), timeout_s=420)
    assert results == {0: "ok", 1: "ok"}


def _rank_main_hsdp(rank, world, cfg_path):
    from modalities_amd.main import Main
    main_obj = Main(Path(cfg_path), experiment_id="hsdp_cfg")
    main_obj.run(main_obj.build_components())
    return "ok"


def test_config_driven_hsdp_world4(tmp_path):
    """World-4 HSDP from YAML: replicate2 x shard2 mesh degrees."""
    rng = np.random.default_rng(41)
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [rng.integers(0, 256, size=600, dtype=np.uint8) for _ in range(10)]
    pbin = tmp_path / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)

    template = Path(__file__).parent / "configs" / "config_tiny_e2e_hsdp.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(tmp_path / "ckpt"))
    text = text.replace("RESULTS_PATH_PLACEHOLDER",
                        str(tmp_path / "evaluation_results.jsonl"))
    cfg = tmp_path / "config.yaml"
    cfg.write_text(text)

    results = run_distributed(_rank_main_hsdp, world_size=4, port=29501,
                              args=(str(cfg),), timeout_s=420)
    assert all(v == "ok" for v in results.values()) and len(results) == 4
    # persisted layout = the SHARD group (2 shards), written once: the
    # replicate peers are write-gated
    ckpt_root = tmp_path / "ckpt" / "hsdp_cfg"
    folder = next(p for p in ckpt_root.iterdir() if p.is_dir())
    names = sorted(f.name for f in folder.iterdir())
    assert names == ["meta.json", "shards_rank_0.pt", "shards_rank_1.pt"], names
    meta = json.loads((folder / "meta.json").read_text())
    assert meta["world_size"] == 2


def _rank_warmstart_hsdp(rank, world, cfg_path, ws_cfg_path):
    from modalities_amd.main import Main

    main_a = Main(Path(cfg_path), experiment_id="hsA")
    main_a.run(main_a.build_components())

    root = Path(cfg_path).parent / "ckpt" / "hsA"
    step4 = [p for p in root.iterdir() if "seen_steps_4" in p.name][0]

    def ws_resolver(key):
        return {"checkpoint_folder_path": str(step4)}[key]

    main_b = Main(Path(ws_cfg_path), experiment_id="hsB",
                  additional_resolver_funs={"warmstart_env": ws_resolver})
    main_b.run(main_b.build_components())
    return "ok"


def test_config_driven_hsdp_warmstart(tmp_path):
    """Resume an HSDP run from its shard-group checkpoint: every replicate
    peer reloads the same shard files through the mesh-aware app_state."""
    import yaml

    rng = np.random.default_rng(51)
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [rng.integers(0, 256, size=600, dtype=np.uint8) for _ in range(10)]
    pbin = tmp_path / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)

    template = Path(__file__).parent / "configs" / "config_tiny_e2e_hsdp.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(tmp_path / "ckpt"))
    text = text.replace("RESULTS_PATH_PLACEHOLDER",
                        str(tmp_path / "a_results.jsonl"))
    cfg = tmp_path / "config.yaml"
    cfg.write_text(text)

    cfg_dict = yaml.safe_load(text.replace(str(tmp_path / "a_results.jsonl"),
                                           str(tmp_path / "b_results.jsonl")))
    cfg_dict["app_state"] = {
        "component_key": "app_state", "variant_key": "sharded_warmstart",
        "config": {
            "model": {"instance_key": "wrapped_model", "pass_type": "BY_REFERENCE"},
            "optimizer": {"instance_key": "optimizer", "pass_type": "BY_REFERENCE"},
            "lr_scheduler": {"instance_key": "scheduler", "pass_type": "BY_REFERENCE"},
            "checkpoint_folder_path": "${warmstart_env:checkpoint_folder_path}",
            "device_mesh": {"instance_key": "device_mesh",
                            "pass_type": "BY_REFERENCE"},
        },
    }
    ws_cfg = tmp_path / "ws.yaml"
    ws_cfg.write_text(yaml.safe_dump(cfg_dict, sort_keys=False))

    results = run_distributed(_rank_warmstart_hsdp, world_size=4, port=29507,
                              args=(str(cfg), str(ws_cfg)), timeout_s=420)
    assert all(v == "ok" for v in results.values()) and len(results) == 4


def _rank_warmstart_cp(rank, world, cfg_path, ws_cfg_path):
    from modalities_amd.main import Main

    main_a = Main(Path(cfg_path), experiment_id="cpA")
    main_a.run(main_a.build_components())
    root = Path(cfg_path).parent / "ckpt" / "cpA"
    step4 = [p for p in root.iterdir() if "seen_steps_4" in p.name][0]

    def ws_resolver(key):
        return {"checkpoint_folder_path": str(step4)}[key]

    main_b = Main(Path(ws_cfg_path), experiment_id="cpB",
                  additional_resolver_funs={"warmstart_env": ws_resolver})
    main_b.run(main_b.build_components())
    return "ok"


def test_config_driven_cp2_warmstart(tmp_path):
    """Resume a CP run: the cp-rank-0 peer wrote the (world-1) layout; both
    CP peers reload it identically through the mesh-aware app_state."""
    import yaml

    rng = np.random.default_rng(61)
    from modalities_amd.dataloader.packed_data import write_pbin
    docs = [rng.integers(0, 256, size=200, dtype=np.uint8) for _ in range(8)]
    pbin = tmp_path / "data.pbin"
    write_pbin(pbin, docs, token_size_in_bytes=1)

    template = Path(__file__).parent / "configs" / "config_tiny_e2e_cp2.yaml"
    text = template.read_text()
    text = text.replace("DATASET_PATH_PLACEHOLDER", str(pbin))
    text = text.replace("CHECKPOINT_DIR_PLACEHOLDER", str(tmp_path / "ckpt"))
    text = text.replace("RESULTS_PATH_PLACEHOLDER",
                        str(tmp_path / "a_results.jsonl"))
    cfg = tmp_path / "config.yaml"
    cfg.write_text(text)

    cfg_dict = yaml.safe_load(text.replace(str(tmp_path / "a_results.jsonl"),
                                           str(tmp_path / "b_results.jsonl")))
    cfg_dict["app_state"] = {
        "component_key": "app_state", "variant_key": "sharded_warmstart",
        "config": {
            "model": {"instance_key": "wrapped_model", "pass_type": "BY_REFERENCE"},
            "optimizer": {"instance_key": "optimizer", "pass_type": "BY_REFERENCE"},
            "lr_scheduler": {"instance_key": "scheduler", "pass_type": "BY_REFERENCE"},
            "checkpoint_folder_path": "${warmstart_env:checkpoint_folder_path}",
            "device_mesh": {"instance_key": "device_mesh",
                            "pass_type": "BY_REFERENCE"},
        },
    }
    ws_cfg = tmp_path / "ws.yaml"
    ws_cfg.write_text(yaml.safe_dump(cfg_dict, sort_keys=False))

    results = run_distributed(_rank_warmstart_cp, world_size=2, port=29513,
                              args=(str(cfg), str(ws_cfg)), timeout_s=420)
    assert results == {0: "ok", 1: "ok"}
    # only the cp-rank-0 peer wrote: flat namespace, single shard, world 1
    root = tmp_path / "ckpt" / "cpA"
    folder = next(p for p in root.iterdir() if p.is_dir())
    names = sorted(f.name for f in folder.iterdir())
    assert names == ["meta.json", "shards_rank_0.pt"], names
