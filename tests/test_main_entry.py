"""End-to-end entry point tests on CPU (single process, dummy data)."""

import os

import torch

import main_training_llama
import main_training_mamba


def test_main_llama_smoke(tmp_path):
    main_training_llama.main(
        model_variant="llama2_125m", use_dummy_dataset=True, batch_size=1,
        seq_length=128, num_steps=3, report_interval=1,
        checkpoint_interval=2, mixed_precision=False,
        ckpt_save_path=str(tmp_path), ckpt_load_path=str(tmp_path),
        vocab_size=256, learning_rate=1e-4, sharding_strategy="fsdp")
    # a checkpoint was written at step 2
    assert os.path.exists(tmp_path / "checkpoints" / "step_2_ckp" / "metadata.pth")
    # resume: runs steps 3.. from the checkpoint
    main_training_llama.main(
        model_variant="llama2_125m", use_dummy_dataset=True, batch_size=1,
        seq_length=128, num_steps=4, report_interval=1,
        checkpoint_interval=2, mixed_precision=False,
        ckpt_save_path=str(tmp_path), ckpt_load_path=str(tmp_path),
        vocab_size=256, learning_rate=1e-4, sharding_strategy="fsdp")
    assert os.path.exists(tmp_path / "checkpoints" / "step_4_ckp" / "metadata.pth")


def test_main_llama_selective_ac(tmp_path):
    main_training_llama.main(
        model_variant="llama2_125m", use_dummy_dataset=True, batch_size=1,
        seq_length=128, num_steps=2, report_interval=1,
        checkpoint_interval=100, mixed_precision=False,
        fsdp_activation_checkpointing=True, selective_checkpointing="1/2",
        ckpt_save_path=str(tmp_path), ckpt_load_path=str(tmp_path),
        vocab_size=256, sharding_strategy="fsdp")


def test_main_mamba_smoke(tmp_path):
    main_training_mamba.main(
        model_variant="mamba_test", use_dummy_dataset=True, batch_size=1,
        seq_length=256, num_steps=2, report_interval=1,
        checkpoint_interval=100, mixed_precision=False,
        ckpt_save_path=str(tmp_path), ckpt_load_path=str(tmp_path),
        learning_rate=1e-4, sharding_strategy="fsdp")
