"""Heimdall fine-tuning: LoRA adapters, trainer, dataset generation and
merged-weight export (reference neural/ — SURVEY.md §2 "Neural training")."""

from .data import InstructionDataset, generate_dataset_from_db
from .export import export_merged, load_merged
from .lora import LoRALinear, inject_lora, lora_state_dict, merge_lora
from .trainer import LoRATrainer, TrainConfig

__all__ = ["LoRALinear", "inject_lora", "merge_lora", "lora_state_dict",
           "InstructionDataset", "generate_dataset_from_db",
           "LoRATrainer", "TrainConfig", "export_merged", "load_merged"]
