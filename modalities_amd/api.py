"""Programmatic API mirroring the CLI (capability parity with reference
src/modalities/api.py:31-402)."""

from enum import Enum
from pathlib import Path
from typing import Callable, Optional

import numpy as np


class FileExistencePolicy(str, Enum):
    SKIP = "skip"
    ERROR = "error"
    OVERRIDE = "override"


def _enforce_file_existence_policy(path: Path, policy: FileExistencePolicy) -> bool:
    """Returns True if processing should be skipped."""
    path = Path(path)
    if not path.exists():
        return False
    if policy == FileExistencePolicy.SKIP:
        return True
    if policy == FileExistencePolicy.OVERRIDE:
        if path.is_file():
            path.unlink()
        return False
    raise FileExistsError(f"{path} already exists (policy=error)")


def create_raw_data_index(src_path: Path, index_path: Optional[Path] = None,
                          file_existence_policy: FileExistencePolicy =
                          FileExistencePolicy.ERROR) -> Optional[int]:
    """Byte-offset index of a JSONL corpus (reference api.py:64-98)."""
    from modalities_amd.dataloader.create_index import (IndexGenerator,
                                                        LargeFileLinesReader)
    src_path = Path(src_path)
    index_path = Path(index_path) if index_path else \
        LargeFileLinesReader.default_index_path(src_path)
    if _enforce_file_existence_policy(index_path, file_existence_policy):
        return None
    return IndexGenerator(src_path).create_index(index_path)


def pack_encoded_data(config_dict: dict,
                      file_existence_policy: FileExistencePolicy =
                      FileExistencePolicy.ERROR) -> Optional[int]:
    """Tokenize + pack a jsonl corpus into .pbin, driven by a config dict
    with keys settings.{src_path,index_path,dst_path,jq_pattern(optional
    json field), eod_token}, tokenizer (component config)
    (reference api.py / create_packed_data.py)."""
    import json

    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.dataloader.create_index import LargeFileLinesReader
    from modalities_amd.dataloader.packed_data import PackedDataGenerator
    from modalities_amd.registry.components import get_default_registry

    settings = config_dict["settings"]
    dst_path = Path(settings["dst_path"])
    if _enforce_file_existence_policy(dst_path, file_existence_policy):
        return None
    factory = ComponentFactory(get_default_registry())
    tokenizer = factory.build_component_by_key(config_dict, "tokenizer")
    field = settings.get("jq_pattern", ".text").lstrip(".")
    eod_token = settings.get("eod_token", "<|endoftext|>")
    eod_id = tokenizer.get_token_id(eod_token)

    reader = LargeFileLinesReader(Path(settings["src_path"]),
                                  Path(settings["index_path"])
                                  if settings.get("index_path") else None)

    def texts():
        for i in range(len(reader)):
            try:
                obj = json.loads(reader[i])
                yield i, obj[field] if isinstance(obj, dict) else str(obj)
            except json.JSONDecodeError:
                yield i, reader[i]

    gen = PackedDataGenerator(
        texts(),
        tokenize_fn_factory=lambda: tokenizer.tokenize,
        eod_token_id=eod_id, vocab_size=tokenizer.vocab_size,
        num_processes=int(settings.get("num_cpus", 1)))
    dst_path.parent.mkdir(parents=True, exist_ok=True)
    return gen.run(dst_path)


def merge_packed_data_files(src_paths: list[Path], target_path: Path) -> None:
    from modalities_amd.dataloader.packed_data import join_embedded_stream_data
    join_embedded_stream_data([Path(p) for p in src_paths], Path(target_path))


def shuffle_tokenized_data(input_data_path: Path, output_data_path: Path,
                           batch_size: int = 1024, seed: Optional[int] = None,
                           file_existence_policy: FileExistencePolicy =
                           FileExistencePolicy.ERROR) -> None:
    from modalities_amd.preprocessing import shuffle_data
    if _enforce_file_existence_policy(Path(output_data_path), file_existence_policy):
        return
    shuffle_data.shuffle_tokenized_data(Path(input_data_path),
                                        Path(output_data_path), batch_size, seed)


def shuffle_jsonl_data(input_data_path: Path, output_data_path: Path,
                       seed: Optional[int] = None,
                       file_existence_policy: FileExistencePolicy =
                       FileExistencePolicy.ERROR) -> None:
    from modalities_amd.preprocessing import shuffle_data
    if _enforce_file_existence_policy(Path(output_data_path), file_existence_policy):
        return
    shuffle_data.shuffle_jsonl_data(Path(input_data_path), Path(output_data_path), seed)


def create_shuffled_dataset_chunk(file_path_list: list[Path],
                                  output_chunk_file_path: Path, chunk_id: int,
                                  num_chunks: int, global_seed: Optional[int] = None,
                                  file_existence_policy: FileExistencePolicy =
                                  FileExistencePolicy.ERROR) -> None:
    from modalities_amd.preprocessing import shuffle_data
    if _enforce_file_existence_policy(Path(output_chunk_file_path),
                                      file_existence_policy):
        return
    shuffle_data.create_shuffled_dataset_chunk(
        [Path(p) for p in file_path_list], Path(output_chunk_file_path),
        chunk_id, num_chunks, global_seed)


def create_shuffled_jsonl_dataset_chunk(file_path_list: list[Path],
                                        output_chunk_file_path: Path, chunk_id: int,
                                        num_chunks: int,
                                        global_seed: Optional[int] = None,
                                        file_existence_policy: FileExistencePolicy =
                                        FileExistencePolicy.ERROR) -> None:
    from modalities_amd.preprocessing import shuffle_data
    if _enforce_file_existence_policy(Path(output_chunk_file_path),
                                      file_existence_policy):
        return
    shuffle_data.create_shuffled_jsonl_dataset_chunk(
        [Path(p) for p in file_path_list], Path(output_chunk_file_path),
        chunk_id, num_chunks, global_seed)


def create_filtered_tokenized_dataset(input_data_path: Path,
                                      output_data_path: Path,
                                      filter_routine: Callable[[int, np.ndarray], bool],
                                      file_existence_policy: FileExistencePolicy =
                                      FileExistencePolicy.ERROR) -> Optional[int]:
    from modalities_amd.preprocessing import shuffle_data
    if _enforce_file_existence_policy(Path(output_data_path), file_existence_policy):
        return None
    return shuffle_data.create_filtered_tokenized_dataset(
        Path(input_data_path), Path(output_data_path), filter_routine)


def _build_text_inference_component(config_path: Path):
    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.config.instantiation_models import \
        TextGenerationInstantiationModel
    from modalities_amd.config.yaml_loader import load_app_config_dict
    from modalities_amd.inference.text_generation import TextInferenceComponent
    from modalities_amd.registry.components import get_default_registry

    config_dict = load_app_config_dict(Path(config_path))
    factory = ComponentFactory(get_default_registry())
    components = factory.build_components(config_dict,
                                          TextGenerationInstantiationModel)
    import torch
    settings = components.settings
    gen_cfg = config_dict.get("text_inference", {})
    comp = TextInferenceComponent(
        components.model, components.tokenizer,
        prompt_template=gen_cfg.get("prompt_template", "{text}"),
        sequence_length=settings.sequence_length,
        temperature=gen_cfg.get("temperature", 1.0),
        top_k=gen_cfg.get("top_k"), top_p=gen_cfg.get("top_p"),
        eod_token=gen_cfg.get("eod_token", "<eod>"),
        device=torch.device(settings.device),
        sample_key=settings.referencing_keys.get("sample_key", "input_ids"),
        prediction_key=settings.referencing_keys.get("prediction_key", "logits"))
    return comp


def generate_text(config_path: Path) -> None:
    """Interactive text generation from a config naming model+tokenizer
    (reference api.py:101-107)."""
    _build_text_inference_component(config_path).run()


def serve(config_path: Path, host: str = "127.0.0.1", port: int = 8000) -> None:
    """HTTP inference server (POST /generate) over the KV-cache decoder."""
    from modalities_amd.inference.server import serve as _serve
    _serve(_build_text_inference_component(config_path), host=host, port=port)


def convert_pytorch_to_hf_checkpoint(config_path: Path, output_hf_checkpoint_dir: Path,
                                     prediction_key: str = "logits"):
    from modalities_amd.conversion.convert_gpt2 import convert_gpt2_to_hf
    return convert_gpt2_to_hf(Path(config_path), Path(output_hf_checkpoint_dir),
                              prediction_key)


def convert_sharded_checkpoint_to_full(checkpoint_folder_path: Path,
                                       config_path: Path,
                                       output_path: Path) -> None:
    """Reassemble a sharded training checkpoint into a single full fp32
    state-dict file loadable by the `model`/`checkpointed` component."""
    import torch

    from modalities_amd.checkpointing.loading import \
        load_full_model_state_from_checkpoint
    from modalities_amd.config.component_factory import ComponentFactory
    from modalities_amd.config.yaml_loader import load_app_config_dict
    from modalities_amd.registry.components import get_default_registry

    config_dict = load_app_config_dict(Path(config_path))
    factory = ComponentFactory(get_default_registry())
    model = factory.build_component_by_key(config_dict, "model")
    load_full_model_state_from_checkpoint(Path(checkpoint_folder_path), model)
    torch.save(model.state_dict(), Path(output_path))
