"""CLI entrypoints (capability parity with reference
src/modalities/__main__.py:44-749): ``run``, ``warmstart``, ``generate_text``,
``convert_pytorch_to_hf_checkpoint``, the ``data ...`` preprocessing family,
``benchmark ...`` sweep helpers and ``profile`` — plus per-rank structured
JSON error logs on any exception."""

import json
import os
import traceback
from datetime import datetime
from pathlib import Path

import click

from modalities_amd import api
from modalities_amd.api import FileExistencePolicy


def _exception_handling(run_fn, error_log_dir: Path = Path("logs")):
    """Per-rank structured JSON error log (reference __main__.py:726-749)."""
    try:
        run_fn()
    except Exception as e:
        rank = os.environ.get("RANK", "0")
        error_log_dir.mkdir(parents=True, exist_ok=True)
        record = {
            "rank": rank,
            "hostname": os.uname().nodename,
            "timestamp": datetime.now().isoformat(),
            "exception_type": type(e).__name__,
            "message": str(e),
            "traceback": traceback.format_exc(),
        }
        with open(error_log_dir / f"error_rank_{rank}.json", "w") as f:
            json.dump(record, f, indent=1)
        raise


@click.group(name="modalities-amd")
def main():
    pass


@main.command(name="run")
@click.option("--config_file_path", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--test_comm", is_flag=True, default=False)
def CMD_entry_point_run_modalities(config_file_path: Path, test_comm: bool):
    from modalities_amd.main import Main
    from modalities_amd.running_env import DistEnv
    from modalities_amd.utils.communication_test import run_communication_test

    def run():
        with DistEnv():
            if test_comm:
                run_communication_test()
            main_obj = Main(config_file_path)
            components = main_obj.build_components()
            main_obj.run(components)

    _exception_handling(run)


@main.command(name="warmstart")
@click.option("--config_file_path", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--last_checkpoint_info_file_path",
              type=click.Path(exists=True, path_type=Path), required=True)
def CMD_entry_point_warmstart_modalities(config_file_path: Path,
                                         last_checkpoint_info_file_path: Path):
    """Resume from the checkpoint pointed to by last_checkpoint_info.json
    (reference __main__.py:137-163: registers the ${warmstart_env:...}
    resolver with the checkpoint path)."""
    from modalities_amd.main import Main
    from modalities_amd.running_env import DistEnv

    with open(last_checkpoint_info_file_path) as f:
        checkpoint_paths = json.load(f)

    def warmstart_env_resolver(key: str):
        if key == "checkpoint_folder_path":
            return checkpoint_paths["checkpoint_folder_path"]
        return checkpoint_paths[key]

    def run():
        with DistEnv():
            main_obj = Main(config_file_path,
                            additional_resolver_funs={
                                "warmstart_env": warmstart_env_resolver})
            components = main_obj.build_components()
            main_obj.run(components)

    _exception_handling(run)


@main.command(name="generate_text")
@click.option("--config_file_path", type=click.Path(exists=True, path_type=Path),
              required=True)
def CMD_entry_point_generate_text(config_file_path: Path):
    api.generate_text(config_file_path)


@main.command(name="serve")
@click.option("--config_file_path", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--host", type=str, default="127.0.0.1")
@click.option("--port", type=int, default=8000)
def CMD_entry_point_serve(config_file_path: Path, host: str, port: int):
    """HTTP inference server (POST /generate) over the KV-cache decoder."""
    api.serve(config_file_path, host=host, port=port)


@main.command(name="convert_pytorch_to_hf_checkpoint")
@click.option("--config_file_path", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--output_hf_checkpoint_dir", type=click.Path(path_type=Path),
              required=True)
@click.option("--prediction_key", type=str, default="logits")
def CMD_entry_point_convert_pytorch_to_hf_checkpoint(
        config_file_path: Path, output_hf_checkpoint_dir: Path, prediction_key: str):
    api.convert_pytorch_to_hf_checkpoint(config_file_path,
                                         output_hf_checkpoint_dir, prediction_key)


@main.command(name="convert_checkpoint_to_full")
@click.option("--checkpoint_folder_path", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--config_file_path", type=click.Path(exists=True, path_type=Path),
              required=True, help="config containing the `model` component")
@click.option("--output_path", type=click.Path(path_type=Path), required=True)
def CMD_convert_checkpoint_to_full(checkpoint_folder_path, config_file_path,
                                   output_path):
    """Reassemble a sharded training checkpoint into one full state-dict
    file (for text generation / HF export)."""
    api.convert_sharded_checkpoint_to_full(checkpoint_folder_path,
                                           config_file_path, output_path)
    click.echo(f"wrote {output_path}")


# ---- data subcommands -------------------------------------------------------

@main.group(name="data")
def data():
    pass


_POLICY = click.Choice([p.value for p in FileExistencePolicy])


@data.command(name="create_raw_index")
@click.argument("src_path", type=click.Path(exists=True, path_type=Path))
@click.option("--index_path", type=click.Path(path_type=Path), default=None)
@click.option("--file_existence_policy", type=_POLICY, default="error")
def CMD_create_raw_index(src_path, index_path, file_existence_policy):
    n = api.create_raw_data_index(src_path, index_path,
                                  FileExistencePolicy(file_existence_policy))
    click.echo(f"indexed {n} lines")


@data.command(name="pack_encoded_data")
@click.argument("config_path", type=click.Path(exists=True, path_type=Path))
@click.option("--file_existence_policy", type=_POLICY, default="error")
def CMD_pack_encoded_data(config_path, file_existence_policy):
    from modalities_amd.config.yaml_loader import load_app_config_dict
    config_dict = load_app_config_dict(config_path)
    n = api.pack_encoded_data(config_dict, FileExistencePolicy(file_existence_policy))
    click.echo(f"packed {n} documents")


@data.command(name="merge_packed_data")
@click.argument("src_paths", type=click.Path(exists=True, path_type=Path), nargs=-1)
@click.argument("target_path", type=click.Path(path_type=Path))
def CMD_merge_packed_data(src_paths, target_path):
    api.merge_packed_data_files(list(src_paths), target_path)


@data.command(name="shuffle_tokenized_data")
@click.option("--input_data_path", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--output_data_path", type=click.Path(path_type=Path), required=True)
@click.option("--batch_size", type=int, default=1024)
@click.option("--seed", type=int, default=None)
@click.option("--file_existence_policy", type=_POLICY, default="error")
def CMD_shuffle_tokenized_data(input_data_path, output_data_path, batch_size, seed,
                               file_existence_policy):
    api.shuffle_tokenized_data(input_data_path, output_data_path, batch_size, seed,
                               FileExistencePolicy(file_existence_policy))


@data.command(name="shuffle_jsonl_data")
@click.option("--input_data_path", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--output_data_path", type=click.Path(path_type=Path), required=True)
@click.option("--seed", type=int, default=None)
@click.option("--file_existence_policy", type=_POLICY, default="error")
def CMD_shuffle_jsonl_data(input_data_path, output_data_path, seed,
                           file_existence_policy):
    api.shuffle_jsonl_data(input_data_path, output_data_path, seed,
                           FileExistencePolicy(file_existence_policy))


@data.command(name="create_shuffled_dataset_chunk")
@click.option("--input_file_list", type=click.Path(exists=True, path_type=Path),
              required=True, help="text file: one pbin path per line")
@click.option("--output_chunk_file_path", type=click.Path(path_type=Path),
              required=True)
@click.option("--chunk_id", type=int, required=True)
@click.option("--num_chunks", type=int, required=True)
@click.option("--global_seed", type=int, default=None)
@click.option("--file_existence_policy", type=_POLICY, default="error")
def CMD_create_shuffled_dataset_chunk(input_file_list, output_chunk_file_path,
                                      chunk_id, num_chunks, global_seed,
                                      file_existence_policy):
    with open(input_file_list) as f:
        files = [Path(line.strip()) for line in f if line.strip()]
    api.create_shuffled_dataset_chunk(files, output_chunk_file_path, chunk_id,
                                      num_chunks, global_seed,
                                      FileExistencePolicy(file_existence_policy))


@data.command(name="create_shuffled_jsonl_chunk")
@click.option("--input_file_list", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--output_chunk_file_path", type=click.Path(path_type=Path),
              required=True)
@click.option("--chunk_id", type=int, required=True)
@click.option("--num_chunks", type=int, required=True)
@click.option("--global_seed", type=int, default=None)
@click.option("--file_existence_policy", type=_POLICY, default="error")
def CMD_create_shuffled_jsonl_chunk(input_file_list, output_chunk_file_path,
                                    chunk_id, num_chunks, global_seed,
                                    file_existence_policy):
    with open(input_file_list) as f:
        files = [Path(line.strip()) for line in f if line.strip()]
    api.create_shuffled_jsonl_dataset_chunk(files, output_chunk_file_path, chunk_id,
                                            num_chunks, global_seed,
                                            FileExistencePolicy(file_existence_policy))


@data.command(name="prepare_instruction_tuning_data")
@click.option("--config_file_path", type=click.Path(exists=True, path_type=Path),
              required=True)
def CMD_prepare_instruction_tuning_data(config_file_path):
    from modalities_amd.dataloader.instruction_tuning import \
        create_instruction_tuning_data
    create_instruction_tuning_data(config_file_path)


# ---- profile ---------------------------------------------------------------

@main.command(name="profile")
@click.option("--config_file_path", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--num_steps", type=int, default=8)
@click.option("--output_dir", type=click.Path(path_type=Path),
              default=Path("profile_out"))
@click.option("--profiler_variant", type=click.Choice(["kernel", "memory",
                                                       "combined"]),
              default="kernel")
def CMD_profile(config_file_path: Path, num_steps: int, output_dir: Path,
                profiler_variant: str):
    """Standalone profiling harness: fwd+bwd+optim steps of the configured
    model on synthetic batches under a steppable profiler (reference
    `modalities profile distributed`, utils/profilers/modalities_profiler.py)."""
    import torch

    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.config.yaml_loader import load_app_config_dict
    from modalities_amd.loss_functions import CLMCrossEntropyLoss
    from modalities_amd.optimizers.optimizer_factory import get_adam_w
    from modalities_amd.registry.components import get_default_registry
    from modalities_amd.utils.profilers import (RandomDatasetBatchGenerator,
                                                SteppableForwardPass,
                                                get_profiler)

    config_dict = load_app_config_dict(config_file_path)
    factory = ComponentFactory(get_default_registry())
    model = factory.build_component_by_key(config_dict, "model")
    settings = config_dict.get("profiling", {})
    device = torch.device("cuda", 0) if torch.cuda.is_available() \
        else torch.device("cpu")
    model = model.to(device)
    opt = get_adam_w(model, lr=settings.get("lr", 1e-4))
    gen = RandomDatasetBatchGenerator(
        vocab_size=model.config.vocab_size,
        sequence_length=settings.get("sequence_length",
                                     model.config.sequence_length),
        batch_size=settings.get("batch_size", 1))
    loss_fn = CLMCrossEntropyLoss("target_ids", "logits")
    stepper = SteppableForwardPass(model, opt, loss_fn, gen, device=device)
    prof = get_profiler(profiler_variant, output_dir)
    with prof:
        for i in range(num_steps):
            loss = stepper.run_step()
            prof.step()
    click.echo(f"profiled {num_steps} steps; last loss "
               f"{loss.item():.4f}; artifacts in {output_dir}")


# ---- benchmark subcommands --------------------------------------------------

@main.group(name="benchmark")
def benchmark():
    pass


@benchmark.command(name="prepare_sweep_configs")
@click.option("--sweep_config_path", type=click.Path(exists=True, path_type=Path),
              required=True)
@click.option("--output_dir", type=click.Path(path_type=Path), required=True)
def CMD_prepare_sweep_configs(sweep_config_path, output_dir):
    from modalities_amd.utils.benchmarking import prepare_sweep_configs
    n = prepare_sweep_configs(sweep_config_path, output_dir)
    click.echo(f"wrote {n} sweep configs")


@benchmark.command(name="list_remaining_runs")
@click.option("--sweep_dir", type=click.Path(exists=True, path_type=Path),
              required=True)
def CMD_list_remaining_runs(sweep_dir):
    from modalities_amd.utils.benchmarking import list_remaining_runs
    for p in list_remaining_runs(sweep_dir):
        click.echo(str(p))


if __name__ == "__main__":
    main()
