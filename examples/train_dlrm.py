"""Golden training example: DLRM on synthetic Criteo-shaped data under
DistributedModelParallel + TrainPipelineSparseDist.

Reference parity: torchrec examples/golden_training/train_dlrm.py.

Launch (1 rank per GPU over RCCL):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/train_dlrm.py
Single process (CPU or one GPU) also works directly.
"""

import os

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from torchrec_amd.datasets.random import RandomRecDataset
from torchrec_amd.distributed.embeddingbag import EmbeddingBagCollectionSharder
from torchrec_amd.distributed.model_parallel import DistributedModelParallel
from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
from torchrec_amd.distributed.planner.types import Topology
from torchrec_amd.distributed.train_pipeline import TrainPipelineSparseDist
from torchrec_amd.distributed.types import ShardingEnv
from torchrec_amd.metrics.metric_module import RecMetricModule, ThroughputMetric
from torchrec_amd.metrics.rec_metric import AUCMetric, NEMetric, RecTaskInfo
from torchrec_amd.models.dlrm import DLRM, DLRMTrain
from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection


def main(num_steps: int = 100, batch_size: int = 256) -> None:
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    use_gpu = torch.cuda.is_available()
    if world > 1:
        dist.init_process_group("nccl" if use_gpu else "gloo")
        env = ShardingEnv.from_process_group(dist.group.WORLD)
        pg = dist.group.WORLD
    else:
        env = ShardingEnv.from_local(1, 0)
        pg = None
    device = torch.device("cuda", rank % 8) if use_gpu else torch.device("cpu")
    if use_gpu:
        torch.cuda.set_device(device)

    num_features = 8
    hash_sizes = [2_000] * num_features  # small tables: ids repeat, the demo converges in ~100 steps
    keys = [f"cat_{i}" for i in range(num_features)]
    tables = [
        EmbeddingBagConfig(
            num_embeddings=hs, embedding_dim=64, name=f"t_{k}", feature_names=[k]
        )
        for k, hs in zip(keys, hash_sizes)
    ]
    model = DLRMTrain(
        DLRM(
            embedding_bag_collection=EmbeddingBagCollection(
                tables=tables, device=torch.device("meta") if use_gpu else None
            ),
            dense_in_features=13,
            dense_arch_layer_sizes=[128, 64],
            over_arch_layer_sizes=[128, 64, 1],
        )
    )
    sharder = EmbeddingBagCollectionSharder(
        fused_params={"optimizer": "rowwise_adagrad", "learning_rate": 0.02}
    )
    planner = EmbeddingShardingPlanner(
        topology=Topology(
            world_size=world,
            compute_device="cuda" if use_gpu else "cpu",
            batch_size=batch_size,
            hbm_cap=None if use_gpu else 1 << 40,
        )
    )
    plan = planner.collective_plan(model, [sharder], pg)
    dmp = DistributedModelParallel(
        model, env=env, plan=plan, sharders=[sharder], device=device
    )
    opt = torch.optim.SGD([p for p in dmp.parameters() if p.requires_grad], lr=0.02)
    pipeline = TrainPipelineSparseDist(dmp, opt, device)

    metrics = RecMetricModule(
        batch_size=batch_size,
        world_size=world,
        rec_tasks=[RecTaskInfo(name="ctr")],
        rec_metrics=[NEMetric([RecTaskInfo(name="ctr")]), AUCMetric([RecTaskInfo(name="ctr")])],
        throughput_metric=ThroughputMetric(batch_size=batch_size, world_size=world),
        compute_interval_steps=50,
    )

    dataset = RandomRecDataset(
        learnable_labels=True,  # label = f(ids): the loop can actually converge
        keys=keys, batch_size=batch_size, hash_sizes=hash_sizes,
        ids_per_feature=1, num_dense=13, seed=100 + rank,
    )
    it = iter(dataset)
    for step in range(num_steps):
        out = pipeline.progress(it)
        loss, logits, labels = out
        metrics.update(
            predictions={"ctr": torch.sigmoid(logits.float().cpu())},
            labels={"ctr": labels.float().cpu()},
        )
        if metrics.should_compute() and rank == 0:
            vals = {k: float(v) for k, v in metrics.compute().items()}
            print(f"step {step}: loss={float(loss):.4f} "
                  f"ne={vals.get('ne-ctr|window_ne', 0):.4f} "
                  f"auc={vals.get('auc-ctr|lifetime_auc', 0):.4f}")
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
