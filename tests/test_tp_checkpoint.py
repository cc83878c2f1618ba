"""TP=2 trainer checkpoint: per-rank model_part files + exact resume."""
import argparse
import os

import torch
import torch.nn as nn

from tests.distributed_utils import run_distributed


def _tp_ckpt_worker(rank, world_size, tmp_root):
    import torch.distributed as dist

    from fengshen_amd.parallel.groups import (
        init_distributed,
        initialize_model_parallel,
    )
    # TP groups must exist BEFORE the model is built (parallel layers
    # read the TP world size at construction)
    init_distributed(backend="gloo")
    initialize_model_parallel(tensor_model_parallel_size=2)

    from fengshen_amd import FengshenModule, Trainer
    from fengshen_amd.models.llama.configuration_llama import llama_tiny_config
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    from fengshen_amd.models.model_utils import (
        add_module_args,
        configure_optimizers,
    )

    class Mod(FengshenModule):
        def __init__(self, args):
            super().__init__()
            self.save_hyperparameters(args)
            # TP inits draw from the model-parallel RNG tracker; reseed it
            # so every build in this process gets identical weights
            from fengshen_amd.parallel.random import model_parallel_manual_seed
            model_parallel_manual_seed(11)
            self.model = LlamaForCausalLM(
                llama_tiny_config(torch_dtype="float32")).float()

        def training_step(self, batch, batch_idx):
            loss = self.model(batch["x"], labels=batch["x"]).loss
            self.log("train_loss", loss)
            return loss

        def configure_optimizers(self):
            return configure_optimizers(self)

    parser = argparse.ArgumentParser()
    add_module_args(parser)
    args = parser.parse_args([])
    args.learning_rate = 1e-3
    args.lr_decay_steps = 4

    g = torch.Generator().manual_seed(5)
    data = [{"x": torch.randint(3, 256, (16,), generator=g)}
            for _ in range(8)]

    def loader():
        return torch.utils.data.DataLoader(
            data, batch_size=4, shuffle=False,
            collate_fn=lambda b: {"x": torch.stack([s["x"] for s in b])})

    def fit(max_steps, root, ckpt=None):
        mod = Mod(args)
        tr = Trainer(max_steps=max_steps, precision="fp32", strategy="zero2",
                     tensor_model_parallel_size=2,
                     default_root_dir=os.path.join(tmp_root, root))
        tr.fit(mod, train_dataloaders=loader(), ckpt_path=ckpt)
        return mod, tr

    # uninterrupted 4 steps
    mod_a, _ = fit(4, "a")
    # interrupted: 2 -> save -> resume to 4
    mod_b, tr_b = fit(2, "b")
    ckpt = os.path.join(tmp_root, "ck")
    tr_b.save_checkpoint(ckpt)
    dist.barrier()
    if rank == 0:
        names = sorted(os.listdir(ckpt))
        assert "model_part_0.pt" in names and "model_part_1.pt" in names, names
    mod_c, tr_c = fit(4, "c", ckpt=ckpt)
    assert tr_c.global_step == 4

    diff = max((pa - pc).abs().max().item()
               for pa, pc in zip(mod_a.parameters(), mod_c.parameters()))
    dist.destroy_process_group()
    return diff


def test_tp2_trainer_checkpoint_resume(tmp_path):
    diffs = run_distributed(_tp_ckpt_worker, world_size=2,
                            args=(str(tmp_path),), timeout=300)
    for d in diffs:
        assert d < 1e-5, f"TP2 resume diverged: {d}"
