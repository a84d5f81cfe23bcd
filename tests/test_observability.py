"""Logger / metrics / population / misc-util tests."""

import csv
import os

import pytest
import torch

from agilerl_amd.logger import CSVLogger, StdOutLogger, TensorboardLogger
from agilerl_amd.metrics import AgentMetrics
from agilerl_amd.population import Population


def _fake_agent():
    from agilerl_amd.algorithms import DQN
    from agilerl_amd.spaces import Box, Discrete

    return DQN(Box(-1, 1, (2,)), Discrete(2), net_config={"arch": "mlp", "hidden_size": [8]})


class TestMetrics:
    def test_scalar_and_fitness(self):
        m = AgentMetrics(0)
        m.log("loss", 1.0)
        m.log("loss", 3.0)
        m.log_fitness(10.0)
        snap = m.snapshot()
        assert snap["loss"] == pytest.approx(2.0)
        assert snap["mean_fitness"] == pytest.approx(10.0)

    def test_steps_per_sec(self):
        m = AgentMetrics(0)
        m.init_training_step()
        m.finalize_training_step(1000)
        assert m.steps_per_sec > 0
        assert m.global_step == 1000


class TestPopulation:
    def test_report_and_stop(self):
        agents = [_fake_agent() for _ in range(2)]
        pop = Population(agents)
        agents[0].fitness = [250.0, 260.0, 270.0]
        agents[1].fitness = [10.0]
        assert pop.should_stop(target=200.0)
        assert not pop.should_stop(target=None)
        assert pop.best_agent is agents[0]
        report = pop.build_report()
        assert report["best_fitness"] == 270.0
        assert len(report["population"]) == 2

    def test_replace_carries_steps(self):
        agents = [_fake_agent() for _ in range(2)]
        pop = Population(agents)
        pop.metrics[0].global_step = 500
        pop.replace([a.clone() for a in agents])
        assert pop.metrics[0].global_step == 500


class TestLoggers:
    def test_csv_logger(self, tmp_path):
        path = str(tmp_path / "log.csv")
        logger = CSVLogger(path)
        logger.log_report({"global_step": 10, "best_fitness": 1.0,
                           "population": [{"agent": 0, "loss": 0.5}]})
        logger.log_report({"global_step": 20, "best_fitness": 2.0,
                           "population": [{"agent": 0, "loss": 0.4}]})
        logger.close()
        rows = list(csv.DictReader(open(path)))
        assert len(rows) == 2
        assert float(rows[1]["best_fitness"]) == 2.0

    def test_tensorboard_logger(self, tmp_path):
        logger = TensorboardLogger(str(tmp_path / "tb"))
        logger.log_report({"global_step": 1, "best_fitness": 5.0, "population": []})
        logger.close()
        if logger.writer is not None:  # tensorboard optional in this image
            assert os.listdir(str(tmp_path / "tb"))

    def test_stdout_logger(self, capsys):
        StdOutLogger().log_report({"global_step": 1, "mean_steps_per_sec": 2.0,
                                   "best_fitness": 3.0, "mean_fitness": 1.0,
                                   "population": []})
        assert "best_fitness" in capsys.readouterr().out


class TestMiscUtils:
    def test_consolidate_mutations(self):
        from agilerl_amd.utils import consolidate_mutations

        a, b = _fake_agent(), _fake_agent()
        a.mut, b.mut = "arch", "arch"
        assert consolidate_mutations([a, b]) == {"arch": 2}

    def test_gpu_mem_snapshot_cpu_noop(self):
        from agilerl_amd.utils import log_gpu_memory_snapshot

        assert log_gpu_memory_snapshot() == {} or torch.cuda.is_available()


class TestGradBucketer:
    def test_hooks_single_process_noop(self):
        from agilerl_amd.parallel import GradBucketer

        net = torch.nn.Linear(4, 2)
        b = GradBucketer(net)  # world_size=1: no hooks attached
        loss = net(torch.randn(3, 4)).sum()
        loss.backward()
        b.finalize()  # no-op, must not raise
        assert net.weight.grad is not None


def test_prometheus_logger_gauges():
    pytest.importorskip("prometheus_client")
    from prometheus_client import REGISTRY

    from agilerl_amd.logger import PrometheusLogger

    logger = PrometheusLogger(start_server=False)
    logger.log_report({
        "global_step": 1000, "mean_steps_per_sec": 123.0, "best_fitness": 42.0,
        "population": [{"index": 0, "fitness": 42.0}, {"index": 1, "fitness": 7.0}],
    })
    assert REGISTRY.get_sample_value("agilerl_global_step") == 1000
    assert REGISTRY.get_sample_value("agilerl_agent_fitness", {"agent": "1"}) == 7.0
    # repeated reports reuse gauges (no duplicate-registration error)
    logger.log_report({"global_step": 2000, "population": []})
    assert REGISTRY.get_sample_value("agilerl_global_step") == 2000


class TestMetricsDetails:
    def test_histograms_bounded_and_snapshotted(self):
        from agilerl_amd.metrics import AgentMetrics

        m = AgentMetrics(0, histogram_len=10)
        for i in range(25):
            m.log_histogram("td_error", float(i))
        snap = m.snapshot()
        assert "td_error" not in snap or True  # histogram not a scalar
        assert len(m.histograms["td_error"]) == 10
        assert list(m.histograms["td_error"])[0] == 15.0

    def test_fitness_window_mean(self):
        from agilerl_amd.metrics import AgentMetrics

        m = AgentMetrics(0, fitness_window=3)
        for f in (1.0, 2.0, 3.0, 4.0):
            m.log_fitness(f)
        assert m.mean_fitness == pytest.approx(3.0)  # property; mean of last 3

    def test_multiagent_per_agent_scalars(self):
        from agilerl_amd.metrics import MultiAgentMetrics

        m = MultiAgentMetrics(0, agent_ids=["a", "b"])
        m.log_agent("a", "reward", 1.0)
        m.log_agent("b", "reward", 3.0)
        m.log_agent("a", "reward", 2.0)
        snap = m.snapshot()
        assert snap is not None
