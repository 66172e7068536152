"""Metrics numeric-parity + Criteo loader tests (reference: torchrec/metrics
per-metric parity tests; torchrec/datasets/tests)."""

import numpy as np
import torch

from torchrec_amd.datasets.criteo import (
    BinaryCriteoUtils,
    CAT_FEATURE_COUNT,
    InMemoryBinaryCriteoIterDataPipe,
    criteo_kaggle,
)
from torchrec_amd.metrics.metric_module import RecMetricModule, ThroughputMetric
from torchrec_amd.metrics.rec_metric import (
    AUCMetric,
    CalibrationMetric,
    MSEMetric,
    NEMetric,
    RecTaskInfo,
)


def _task():
    return [RecTaskInfo(name="t")]


class TestMetrics:
    def test_ne(self):
        torch.manual_seed(0)
        m = NEMetric(_task())
        p = torch.rand(100)
        y = (torch.rand(100) > 0.5).float()
        m.update(predictions={"t": p}, labels={"t": y})
        ne = m.compute()["ne-t|lifetime_ne"]
        # reference formula
        pc = p.double().clamp(1e-7, 1 - 1e-7)
        ce = -(y * pc.log() + (1 - y) * (1 - pc).log()).mean()
        base_p = y.double().mean().clamp(1e-7, 1 - 1e-7)
        base = -(base_p * base_p.log() + (1 - base_p) * (1 - base_p).log())
        torch.testing.assert_close(ne.reshape(()), ce / base, atol=1e-6, rtol=1e-6)

    def test_auc_perfect_and_random(self):
        m = AUCMetric(_task())
        p = torch.tensor([0.9, 0.8, 0.2, 0.1])
        y = torch.tensor([1.0, 1.0, 0.0, 0.0])
        m.update(predictions={"t": p}, labels={"t": y})
        assert float(m.compute()["auc-t|lifetime_auc"]) == 1.0
        m2 = AUCMetric(_task())
        p2 = torch.tensor([0.1, 0.9, 0.1, 0.9])
        y2 = torch.tensor([1.0, 0.0, 0.0, 1.0])
        m2.update(predictions={"t": p2}, labels={"t": y2})
        assert abs(float(m2.compute()["auc-t|lifetime_auc"]) - 0.5) < 1e-6

    def test_calibration_and_mse(self):
        c = CalibrationMetric(_task())
        mse = MSEMetric(_task())
        p = torch.tensor([0.5, 0.5])
        y = torch.tensor([1.0, 0.0])
        c.update(predictions={"t": p}, labels={"t": y})
        mse.update(predictions={"t": p}, labels={"t": y})
        assert abs(float(c.compute()["calibration-t|lifetime_calibration"]) - 1.0) < 1e-6
        assert abs(float(mse.compute()["mse-t|lifetime_mse"]) - 0.25) < 1e-6

    def test_metric_module(self):
        mm = RecMetricModule(
            batch_size=4,
            world_size=1,
            rec_tasks=_task(),
            rec_metrics=[NEMetric(_task())],
            throughput_metric=ThroughputMetric(batch_size=4, world_size=1),
            compute_interval_steps=2,
        )
        for _ in range(2):
            mm.update(
                predictions={"t": torch.rand(4)}, labels={"t": torch.ones(4)}
            )
        assert mm.should_compute()
        out = mm.compute()
        assert any(k.startswith("ne-") for k in out)
        assert "throughput-throughput|total_examples" in out


class TestCriteo:
    def _write_tsv(self, path, rows=32):
        rng = np.random.default_rng(0)
        with open(path, "w") as f:
            for _ in range(rows):
                label = rng.integers(0, 2)
                dense = "\t".join(str(rng.integers(0, 100)) for _ in range(13))
                sparse = "\t".join(format(rng.integers(0, 1 << 31), "x") for _ in range(26))
                f.write(f"{label}\t{dense}\t{sparse}\n")

    def test_tsv_and_binary_pipeline(self, tmp_path):
        tsv = str(tmp_path / "day_0.tsv")
        self._write_tsv(tsv)
        rows = list(criteo_kaggle(tsv))
        assert len(rows) == 32
        label, dense, sparse = rows[0]
        assert len(dense) == 13 and len(sparse) == 26

        d, s, l = (str(tmp_path / n) for n in ("d.npy", "s.npy", "l.npy"))
        BinaryCriteoUtils.tsv_to_npys(tsv, d, s, l)
        assert BinaryCriteoUtils.get_shape_from_npy(d) == (32, 13)

        dp0 = InMemoryBinaryCriteoIterDataPipe(
            [d], [s], [l], batch_size=8, rank=0, world_size=2, hashes=[1000] * 26
        )
        dp1 = InMemoryBinaryCriteoIterDataPipe(
            [d], [s], [l], batch_size=8, rank=1, world_size=2, hashes=[1000] * 26
        )
        b0 = list(dp0)
        b1 = list(dp1)
        assert len(b0) == 2 and len(b1) == 2
        batch = b0[0]
        assert batch.dense_features.shape == (8, 13)
        assert batch.sparse_features.stride() == 8
        assert len(batch.sparse_features.keys()) == CAT_FEATURE_COUNT
        assert int(batch.sparse_features.values().max()) < 1000


def test_accuracy_precision_recall_gauc():
    import torch

    from torchrec_amd.metrics.rec_metric import (
        AccuracyMetric,
        PrecisionMetric,
        RecallMetric,
        RecTaskInfo,
        grouped_auc,
    )

    preds = torch.tensor([0.9, 0.8, 0.3, 0.2, 0.6, 0.1])
    labels = torch.tensor([1.0, 1.0, 0.0, 1.0, 0.0, 0.0])
    tasks = [RecTaskInfo(name="t")]
    acc = AccuracyMetric(tasks)
    acc.update(predictions={"t": preds}, labels={"t": labels})
    a = acc.compute()["accuracy-t|lifetime_accuracy"]
    # correct: 0.9->1, 0.8->1, 0.3->0, 0.2 vs 1 wrong, 0.6 vs 0 wrong, 0.1->0
    assert abs(float(a) - 4 / 6) < 1e-6
    prec = PrecisionMetric(tasks)
    prec.update(predictions={"t": preds}, labels={"t": labels})
    p = prec.compute()["precision-t|lifetime_precision"]
    assert abs(float(p) - 2 / 3) < 1e-6  # predicted pos: 0.9,0.8,0.6 -> 2 tp
    rec = RecallMetric(tasks)
    rec.update(predictions={"t": preds}, labels={"t": labels})
    r = rec.compute()["recall-t|lifetime_recall"]
    assert abs(float(r) - 2 / 3) < 1e-6  # 3 positives, 2 recovered
    # GAUC: group 0 perfectly ordered, group 1 inverted
    g = torch.tensor([0, 0, 0, 1, 1, 1])
    preds2 = torch.tensor([0.9, 0.8, 0.1, 0.2, 0.9, 0.8])
    labels2 = torch.tensor([1.0, 1.0, 0.0, 1.0, 0.0, 0.0])
    gauc = grouped_auc(preds2, labels2, g)
    assert abs(float(gauc) - 0.5) < 1e-6  # (1.0 + 0.0) / 2


def test_cpu_offloaded_metric_module():
    import torch

    from torchrec_amd.metrics.cpu_offloaded_metric_module import (
        CPUOffloadedRecMetricModule,
    )
    from torchrec_amd.metrics.metric_module import RecMetricModule
    from torchrec_amd.metrics.rec_metric import NEMetric, RecTaskInfo

    tasks = [RecTaskInfo(name="t")]
    off = CPUOffloadedRecMetricModule(
        batch_size=4, world_size=1, rec_tasks=tasks, rec_metrics=[NEMetric(tasks)]
    )
    sync = RecMetricModule(
        batch_size=4, world_size=1, rec_tasks=tasks, rec_metrics=[NEMetric(tasks)]
    )
    g = torch.Generator().manual_seed(0)
    for _ in range(5):
        p = torch.rand(16, generator=g)
        y = (torch.rand(16, generator=g) > 0.5).float()
        off.update(predictions={"t": p}, labels={"t": y})
        sync.update(predictions={"t": p}, labels={"t": y})
    a = off.compute()
    b = sync.compute()
    for k in b:
        torch.testing.assert_close(a[k], b[k])
    off.shutdown()


def test_criteo_shard_math():
    from torchrec_amd.datasets.criteo import BinaryCriteoUtils

    lengths = [100, 50, 75]  # three files, 225 rows
    seen = []
    for rank in range(4):
        ranges, rem = BinaryCriteoUtils.get_file_row_ranges_and_remainder(
            lengths, rank, 4
        )
        n = sum(hi - lo for lo, hi in ranges.values())
        assert n == 225 // 4
        for f, (lo, hi) in ranges.items():
            assert 0 <= lo <= hi <= lengths[f]
            seen.extend((f, r) for r in range(lo, hi))
    # ranks cover disjoint rows
    assert len(seen) == len(set(seen)) == (225 // 4) * 4


class TestExtendedMetrics:
    def _mk(self, cls, **kw):
        from torchrec_amd.metrics.rec_metric import RecTaskInfo

        return cls(tasks=[RecTaskInfo(name="t")], **kw)

    def test_mae_ctr_weighted_avg_logloss(self):
        from torchrec_amd.metrics.extended_metrics import (
            CTRMetric, LogLossMetric, MAEMetric, WeightedAvgMetric,
        )

        torch.manual_seed(0)
        p = torch.rand(100)
        y = (torch.rand(100) > 0.7).double()
        for cls, expect in [
            (MAEMetric, (p.double() - y).abs().mean()),
            (CTRMetric, y.mean()),
            (WeightedAvgMetric, p.double().mean()),
        ]:
            m = self._mk(cls)
            m.update(predictions={"t": p}, labels={"t": y})
            got = m.compute()[f"{cls.NAME}-t|lifetime_{cls.NAME}"]
            torch.testing.assert_close(got.squeeze(), expect, atol=1e-6, rtol=1e-6)
        m = self._mk(LogLossMetric)
        m.update(predictions={"t": p}, labels={"t": y})
        got = m.compute()["logloss-t|lifetime_logloss"]
        pc = p.double().clamp(1e-7, 1 - 1e-7)
        ce = -(y * pc.log() + (1 - y) * (1 - pc).log())
        torch.testing.assert_close(got.squeeze(), ce.mean(), atol=1e-6, rtol=1e-6)

    def test_ndcg_perfect_and_inverted(self):
        from torchrec_amd.metrics.extended_metrics import NDCGMetric

        m = self._mk(NDCGMetric)
        # one session with labels [1, 0]: perfect ranking -> ndcg 1
        m.update(
            predictions={"t": torch.tensor([0.9, 0.1])},
            labels={"t": torch.tensor([1.0, 0.0])},
            session_ids={"t": torch.tensor([7, 7])},
        )
        got = m.compute()["ndcg-t|lifetime_ndcg"]
        torch.testing.assert_close(got.squeeze(), torch.tensor(1.0).double())

        m2 = self._mk(NDCGMetric)
        # inverted ranking: dcg = 1/log2(3), idcg = 1 -> ndcg = 0.6309
        m2.update(
            predictions={"t": torch.tensor([0.1, 0.9])},
            labels={"t": torch.tensor([1.0, 0.0])},
            session_ids={"t": torch.tensor([7, 7])},
        )
        got2 = m2.compute()["ndcg-t|lifetime_ndcg"]
        import math

        torch.testing.assert_close(
            got2.squeeze(), torch.tensor(1.0 / math.log2(3)).double(),
            atol=1e-6, rtol=1e-6,
        )

    def test_recall_session_topk(self):
        from torchrec_amd.metrics.extended_metrics import RecallSessionMetric

        m = self._mk(RecallSessionMetric, top_k=2)
        # session A: positives ranked 1st and 3rd -> 1 of 2 in top-2
        # session B: positive ranked 1st -> 1 of 1
        m.update(
            predictions={"t": torch.tensor([0.9, 0.8, 0.7, 0.95, 0.1])},
            labels={"t": torch.tensor([1.0, 0.0, 1.0, 1.0, 0.0])},
            session_ids={"t": torch.tensor([1, 1, 1, 2, 2])},
        )
        got = m.compute()["recall_session-t|lifetime_recall_session"]
        torch.testing.assert_close(got.squeeze(), torch.tensor(2.0 / 3.0).double())

    def test_fused_tasks_matches_unfused(self):
        from torchrec_amd.metrics.rec_metric import (
            NEMetric, RecComputeMode, RecTaskInfo,
        )

        torch.manual_seed(0)
        tasks = [RecTaskInfo(name="a"), RecTaskInfo(name="b")]
        fused = NEMetric(tasks, compute_mode=RecComputeMode.FUSED_TASKS_COMPUTATION)
        unfused = NEMetric(tasks)
        for seed in range(3):
            g = torch.Generator().manual_seed(seed)
            preds = {t.name: torch.rand(50, generator=g) for t in tasks}
            labels = {t.name: (torch.rand(50, generator=g) > 0.5).double() for t in tasks}
            fused.update(predictions=preds, labels=labels)
            unfused.update(predictions=preds, labels=labels)
        rf, ru = fused.compute(), unfused.compute()
        for k in ru:
            torch.testing.assert_close(rf[k].squeeze(), ru[k].squeeze(),
                                       atol=1e-9, rtol=1e-9)

    def test_tower_qps(self):
        from torchrec_amd.metrics.extended_metrics import TowerQPSMetric

        m = TowerQPSMetric(["over", "user"], warmup_steps=0)
        for _ in range(5):
            m.update({"over": 100, "user": 50})
        out = m.compute()
        assert out["tower_qps-over"] > 0
        assert abs(out["tower_qps-over"] / out["tower_qps-user"] - 2.0) < 1e-6
