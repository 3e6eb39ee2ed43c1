"""T5-base span-corruption pretraining, synthetic data.

Reference recipe: configs/t5_large_pretrain.py (T5 enc-dec, MLM spans).
"""

from libai_amd.config import LazyCall
from libai_amd.data import build_nlp_train_loader
from libai_amd.data.datasets import SyntheticBertDataset

from .common.models.t5 import cfg as t5_cfg  # noqa: F401
from .common.models.t5 import pretrain_model as model
from .common.optim import optim
from .common.train import train


class _SyntheticT5(SyntheticBertDataset):
    """Synthetic encoder/decoder batches shaped like T5Dataset's output."""

    def __getitem__(self, idx):
        import torch

        from libai_amd.data.structures import DistTensorData, Instance

        g = torch.Generator().manual_seed(1234 + idx)
        enc = torch.randint(3, 32000, (512,), generator=g)
        dec = torch.randint(3, 32000, (114,), generator=g)
        return Instance(
            encoder_input_ids=DistTensorData(enc),
            decoder_input_ids=DistTensorData(dec),
            encoder_attn_mask=DistTensorData(torch.ones(512, dtype=torch.uint8)),
            lm_labels=DistTensorData(torch.randint(3, 32000, (114,), generator=g),
                                     placement_idx=-1),
            loss_mask=DistTensorData(torch.ones(114, dtype=torch.long),
                                     placement_idx=-1),
        )


dataloader = dict(
    train=LazyCall(build_nlp_train_loader)(
        dataset=LazyCall(_SyntheticT5)(vocab_size=32128, seq_length=512,
                                       size=65536),
        train_batch_size=16,
        num_workers=2,
    ),
)

train.update(
    output_dir="./output/t5_pretrain",
    train_micro_batch_size=16,
    train_iter=1000,
    log_period=10,
    amp=dict(enabled=True),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=t5_cfg.hidden_layers,
    ),
)
