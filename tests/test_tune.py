"""Ray Tune parity tests: grid/random search, ResultGrid, ASHA early
stopping, checkpoint reporting."""
import pytest


@pytest.fixture(scope="module")
def ray_mod():
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()  # never inherit another module's (possibly dying) session
    if not ray.is_initialized():
        ray.init(num_cpus=8)
    yield ray
    ray.shutdown()


def test_grid_and_random_search(ray_mod, tmp_path_factory):
    from ant_ray_amd import tune
    from ant_ray_amd.train import RunConfig

    def objective(config):
        score = -((config["x"] - 3) ** 2) + config["b"]
        tune.report({"score": score})

    tuner = tune.Tuner(
        objective,
        param_space={"x": tune.grid_search([1, 2, 3, 4]),
                     "b": tune.choice([10])},
        tune_config=tune.TuneConfig(metric="score", mode="max"),
        run_config=RunConfig(name="grid",
                             storage_path=str(tmp_path_factory.mktemp("t"))),
    )
    grid = tuner.fit()
    assert len(grid) == 4
    best = grid.get_best_result(metric="score", mode="max")
    assert best.metrics["config"]["x"] == 3
    assert best.metrics["score"] == 10


def test_num_samples_random(ray_mod):
    from ant_ray_amd import tune

    def objective(config):
        tune.report({"v": config["u"]})

    grid = tune.Tuner(
        objective,
        param_space={"u": tune.uniform(0, 1)},
        tune_config=tune.TuneConfig(num_samples=5, metric="v", seed=42),
    ).fit()
    vals = [r.metrics["v"] for r in grid]
    assert len(vals) == 5 and len(set(vals)) == 5
    assert all(0 <= v <= 1 for v in vals)


def test_asha_stops_bad_trials(ray_mod):
    from ant_ray_amd import tune

    def objective(config):
        import time

        for i in range(20):
            time.sleep(0.05)
            tune.report({"acc": config["q"] * (i + 1)})

    # strong trials run first (concurrency 2) and populate the rungs, so the
    # weak trials get compared -- and stopped -- at their first milestones
    grid = tune.Tuner(
        objective,
        param_space={"q": tune.grid_search([2.0, 1.0, 0.01, 0.02])},
        tune_config=tune.TuneConfig(
            metric="acc", mode="max", max_concurrent_trials=2,
            scheduler=tune.ASHAScheduler(max_t=20, grace_period=2,
                                         reduction_factor=2),
        ),
    ).fit()
    best = grid.get_best_result(metric="acc", mode="max")
    assert best.metrics["config"]["q"] == 2.0
    # at least one weak trial was stopped before max_t
    iters = [r.metrics["training_iteration"] for r in grid]
    assert min(iters) < 20
    assert max(iters) == 20


def test_checkpoint_and_errors(ray_mod, tmp_path_factory):
    from ant_ray_amd import tune
    from ant_ray_amd.train import Checkpoint, RunConfig

    def objective(config):
        import os
        import tempfile

        if config["x"] == 13:
            raise ValueError("unlucky")
        with tempfile.TemporaryDirectory() as d:
            with open(os.path.join(d, "state.txt"), "w") as f:
                f.write(str(config["x"]))
            tune.report({"x2": config["x"] ** 2},
                        checkpoint=Checkpoint.from_directory(d))

    grid = tune.Tuner(
        objective,
        param_space={"x": tune.grid_search([2, 13])},
        tune_config=tune.TuneConfig(metric="x2", mode="max"),
        run_config=RunConfig(name="ck",
                             storage_path=str(tmp_path_factory.mktemp("t"))),
    ).fit()
    assert len(grid.errors) == 1
    ok = [r for r in grid if r.error is None][0]
    import os

    with open(os.path.join(ok.checkpoint.path, "state.txt")) as f:
        assert f.read() == "2"


def test_pbt_exploits_and_perturbs(ray_mod, tmp_path_factory):
    """PopulationBasedTraining: a trial started with a bad hyperparameter
    gets relaunched from a good trial's checkpoint with a mutated config
    (reference tune/schedulers/pbt.py exploit/explore)."""
    import json
    import os

    from ant_ray_amd import tune
    from ant_ray_amd.train import Checkpoint, RunConfig
    from ant_ray_amd.tune import PopulationBasedTraining, Tuner

    storage = str(tmp_path_factory.mktemp("pbt"))

    def trainable(config):
        import tempfile

        from ant_ray_amd import train

        # resume: score continues from the donor's checkpointed score
        base, start_it = 0.0, 0
        ck = train.get_checkpoint()
        if ck is not None:
            with open(os.path.join(ck.path, "state.json")) as f:
                st = json.load(f)
            base, start_it = st["score"], st["it"]
        lr = config["lr"]
        import time as _t

        for it in range(start_it + 1, 31):
            _t.sleep(0.3)  # let the Tuner poll between iterations even
            # when the whole machine is under load (a trial finishing
            # inside ONE poll window can never be perturbed)
            base += lr  # good lr climbs faster
            with tempfile.TemporaryDirectory() as d:
                with open(os.path.join(d, "state.json"), "w") as f:
                    json.dump({"score": base, "it": it, "lr": lr}, f)
                train.report({"score": base, "training_iteration": it,
                              "lr": lr},
                             checkpoint=Checkpoint.from_directory(d))

    pbt = PopulationBasedTraining(
        perturbation_interval=5,
        hyperparam_mutations={"lr": [0.1, 1.0]},
        quantile_fraction=0.5,
        resample_probability=1.0,
        seed=7,
    )
    from ant_ray_amd.tune import TuneConfig
    tuner = Tuner(
        trainable,
        param_space={"lr": tune.grid_search([0.001, 1.0])},
        tune_config=TuneConfig(metric="score", mode="max", scheduler=pbt),
        run_config=RunConfig(name="pbt", storage_path=storage),
    )
    grid = tuner.fit()
    scores = sorted(r.metrics["score"] for r in grid)
    # the bad-lr trial must have been perturbed onto a useful lr: its
    # final score is far above what 20 iterations of lr=0.001 (0.02) give
    assert scores[0] > 1.0, scores
    assert grid.get_best_result(metric="score", mode="max").metrics["score"] >= 29.0


def test_tuner_restore(ray_mod, tmp_path_factory):
    """Tuner.restore resumes an interrupted experiment: finished trials
    keep results, unfinished ones re-run from their checkpoints."""
    import json
    import os

    from ant_ray_amd import tune
    from ant_ray_amd.train import Checkpoint, RunConfig
    from ant_ray_amd.tune import TuneConfig, Tuner

    storage = str(tmp_path_factory.mktemp("resume"))
    poison = os.path.join(storage, "poison")
    open(poison, "w").close()  # first run: trial with x>=2 dies

    def trainable(config):
        import time as _t

        from ant_ray_amd import train

        start = 0
        ck = train.get_checkpoint()
        if ck is not None:
            with open(os.path.join(ck.path, "s.json")) as f:
                start = json.load(f)["it"] + 1
        for it in range(start, 4):
            _t.sleep(0.1)
            if config["x"] >= 2 and os.path.exists(poison):
                raise RuntimeError("interrupted")
            import tempfile

            with tempfile.TemporaryDirectory() as d:
                with open(os.path.join(d, "s.json"), "w") as f:
                    json.dump({"it": it}, f)
                train.report({"it": it, "x": config["x"],
                              "resumed_from": start},
                             checkpoint=Checkpoint.from_directory(d))

    tuner = Tuner(trainable,
                  param_space={"x": tune.grid_search([1, 2])},
                  tune_config=TuneConfig(metric="it", mode="max"),
                  run_config=RunConfig(name="resume_exp",
                                       storage_path=storage))
    grid = tuner.fit()
    assert len(grid.errors) == 1  # the poisoned trial failed

    os.unlink(poison)
    restored = Tuner.restore(os.path.join(storage, "resume_exp"),
                             trainable, resume_errored=True)
    grid2 = restored.fit()
    assert not grid2.errors
    by_x = {r.metrics["x"]: r.metrics for r in grid2}
    assert by_x[1]["it"] == 3          # finished trial carried over
    assert by_x[2]["it"] == 3          # interrupted trial completed


def test_trainable_class_api(ray_mod, tmp_path_factory):
    """Class-API Trainable: setup/step loop with stop criteria, checkpoint
    at end, registry launch by name (parity: trainable/trainable.py,
    tune/registry.py)."""
    import os

    from ant_ray_amd import tune
    from ant_ray_amd.train import RunConfig

    class MyTrainable(tune.Trainable):
        def setup(self, config):
            self.base = config["base"]

        def step(self):
            return {"score": self.base * self.training_iteration}

        def save_checkpoint(self, d):
            with open(os.path.join(d, "state.txt"), "w") as f:
                f.write(str(self.training_iteration))
            return d

    storage = str(tmp_path_factory.mktemp("tune_cls"))
    tuner = tune.Tuner(
        MyTrainable,
        param_space={"base": tune.grid_search([1, 2])},
        run_config=RunConfig(storage_path=storage,
                             stop={"training_iteration": 4}),
    )
    rg = tuner.fit()
    assert len(rg) == 2 and not rg.errors
    best = rg.get_best_result(metric="score", mode="max")
    # step() sees the pre-increment iteration count (reference Trainable
    # numbering): the stopping result (training_iteration=4) computed 2*3
    assert best.metrics["score"] == 2 * 3
    assert best.checkpoint is not None
    with open(os.path.join(best.checkpoint.path, "state.txt")) as f:
        assert f.read() == "4"

    # by-name via registry
    tune.register_trainable("my_trainable", MyTrainable)
    rg2 = tune.Tuner(
        "my_trainable", param_space={"base": [3]},
        run_config=RunConfig(storage_path=storage,
                             stop={"training_iteration": 2}),
    ).fit()
    assert not rg2.errors


def test_stopper_callback_reporter(ray_mod, tmp_path_factory):
    """RunConfig stop dict + Stopper + Callback hooks + CLIReporter
    (parity: tune/stopper, tune/callback.py, progress_reporter.py)."""
    from ant_ray_amd import tune
    from ant_ray_amd.train import RunConfig
    from ant_ray_amd.tune.stopper import MaximumIterationStopper

    def loop(config):
        import time as _t

        for i in range(100):
            tune.report({"loss": 1.0 / (i + 1)})
            _t.sleep(0.05)  # slow enough for the Tuner to stop us in-band

    events = []

    class Rec(tune.Callback):
        def on_trial_start(self, iteration, trials, trial, **info):
            events.append(("start", trial))

        def on_trial_result(self, iteration, trials, trial, result, **info):
            events.append(("result", result["training_iteration"]))

        def on_trial_complete(self, iteration, trials, trial, **info):
            events.append(("complete", trial))

    storage = str(tmp_path_factory.mktemp("tune_stop"))
    rg = tune.Tuner(
        loop, param_space={},
        run_config=RunConfig(
            storage_path=storage,
            stop=MaximumIterationStopper(5),
            callbacks=[Rec()],
            progress_reporter=tune.CLIReporter(max_report_frequency=0.0),
        ),
    ).fit()
    assert not rg.errors
    # stopped well before 100 iterations
    assert rg[0].metrics["training_iteration"] <= 60
    kinds = [e[0] for e in events]
    assert "start" in kinds and "result" in kinds and "complete" in kinds


def test_sampling_domains_and_analysis(ray_mod, tmp_path_factory):
    """Quantized/log domains + sample_from resolve in param spaces;
    ExperimentAnalysis reloads a finished experiment (parity:
    tune/search/sample.py, analysis/experiment_analysis.py)."""
    import os

    from ant_ray_amd import tune
    from ant_ray_amd.train import RunConfig

    def trial(config):
        assert config["q"] in {0.0, 0.25, 0.5, 0.75, 1.0}
        assert 1 <= config["li"] <= 1000 and isinstance(config["li"], int)
        assert config["fixed"] == 42
        tune.report({"obj": config["q"]})

    storage = str(tmp_path_factory.mktemp("tune_dom"))
    rg = tune.Tuner(
        trial,
        param_space={"q": tune.quniform(0, 1, 0.25),
                     "li": tune.lograndint(1, 1000),
                     "fixed": tune.sample_from(lambda spec: 42)},
        tune_config=tune.TuneConfig(num_samples=6, seed=3),
        run_config=RunConfig(storage_path=storage, name="dom_exp"),
    ).fit()
    assert len(rg) == 6 and not rg.errors

    ea = tune.ExperimentAnalysis(os.path.join(storage, "dom_exp"))
    assert len(ea.trials) == 6
    cfg = ea.get_best_config(metric="obj", mode="max")
    assert cfg["fixed"] == 42
    df = ea.dataframe()
    assert len(df) == 6

    # create_scheduler / create_searcher factories
    assert type(tune.create_scheduler("asha")).__name__ == "ASHAScheduler"
    with pytest.raises(ValueError):
        tune.create_scheduler("nope")


def test_bayesopt_searcher_finds_optimum(ray_mod, tmp_path_factory):
    """Native GP-EI searcher drives the Tuner toward the quadratic's
    optimum better than its own random warmup."""
    from ant_ray_amd import tune
    from ant_ray_amd.train import RunConfig
    from ant_ray_amd.tune.search import BayesOptSearch

    def objective(config):
        x, y = config["x"], config["y"]
        tune.report({"score": -((x - 0.7) ** 2) - (y + 0.3) ** 2})

    search = BayesOptSearch(metric="score", mode="max",
                            random_search_steps=6, seed=0)
    tuner = tune.Tuner(
        objective,
        param_space={"x": tune.uniform(-2, 2), "y": tune.uniform(-2, 2)},
        tune_config=tune.TuneConfig(metric="score", mode="max",
                                    num_samples=24, search_alg=search,
                                    max_concurrent_trials=2),
        run_config=RunConfig(name="bo",
                             storage_path=str(tmp_path_factory.mktemp("bo"))),
    )
    res = tuner.fit()
    best = res.get_best_result(metric="score", mode="max")
    assert len(list(res)) == 24
    assert best.metrics["score"] > -0.25, best.metrics
    # the model-based phase must beat the pure-random warmup's best
    warm = sorted((r.metrics["score"] for r in list(res)[:6]), reverse=True)
    assert best.metrics["score"] >= warm[0]


def test_tpe_searcher_and_concurrency_limiter(ray_mod, tmp_path_factory):
    from ant_ray_amd import tune
    from ant_ray_amd.train import RunConfig
    from ant_ray_amd.tune.search import ConcurrencyLimiter, TPESearch

    seen = []

    def objective(config):
        seen.append(config["lr"])
        tune.report({"loss": abs(config["lr"] - 1e-2)})

    search = ConcurrencyLimiter(
        TPESearch(metric="loss", mode="min", n_startup=6, seed=1),
        max_concurrent=2)
    res = tune.Tuner(
        objective,
        param_space={"lr": tune.loguniform(1e-5, 1.0)},
        tune_config=tune.TuneConfig(metric="loss", mode="min",
                                    num_samples=18, search_alg=search,
                                    max_concurrent_trials=4),
        run_config=RunConfig(name="tpe",
                             storage_path=str(tmp_path_factory.mktemp("tpe"))),
    ).fit()
    best = res.get_best_result(metric="loss", mode="min")
    assert best.metrics["loss"] < 0.05


def test_repeater_averages(ray_mod, tmp_path_factory):
    from ant_ray_amd import tune
    from ant_ray_amd.train import RunConfig
    from ant_ray_amd.tune.search import Repeater, TPESearch

    inner = TPESearch(metric="v", mode="max", n_startup=2, seed=2)
    rep = Repeater(inner, repeat=3)

    def objective(config):
        # noisy objective; __trial_index__ marks the repeat index
        assert "__trial_index__" in config
        tune.report({"v": config["x"] + 0.01 * config["__trial_index__"]})

    tune.Tuner(
        objective,
        param_space={"x": tune.uniform(0, 1)},
        tune_config=tune.TuneConfig(metric="v", mode="max", num_samples=9,
                                    search_alg=rep),
        run_config=RunConfig(name="rep",
                             storage_path=str(tmp_path_factory.mktemp("rp"))),
    ).fit()
    # 9 trials = 3 groups of 3; inner searcher observed exactly 3 results
    assert len(inner._y) == 3


def test_searcher_wrapper_paths():
    from ant_ray_amd.tune.search.bayesopt import BayesOptSearch  # noqa
    from ant_ray_amd.tune.search.bohb import TuneBOHB  # noqa
    from ant_ray_amd.tune.search.hyperopt import HyperOptSearch
    from ant_ray_amd.tune.search.optuna import OptunaSearch
    from ant_ray_amd import tune

    s = OptunaSearch(metric="m", mode="min")
    s.set_search_properties("m", "min", {"x": tune.uniform(0, 1)})
    cfg = s.suggest("t1")
    assert 0 <= cfg["x"] <= 1
    s.on_trial_complete("t1", {"m": 0.5})
    h = HyperOptSearch(space={"x": tune.uniform(0, 1)}, metric="m")
    assert 0 <= h.suggest("t2")["x"] <= 1
    assert type(tune.create_searcher("bayesopt")).__name__ == "BayesOptSearch"


def test_hyperband_scheduler_prunes(ray_mod):
    """Protocol-level: brackets assign round-robin; a bad trial arriving
    at a rung AFTER its peers is cut, a good one survives to max_t."""
    from ant_ray_amd.tune.schedulers import (CONTINUE, STOP,
                                             HyperBandScheduler)

    s = HyperBandScheduler(metric="score", mode="max", max_t=27,
                           reduction_factor=3)
    s.set_objective("score", "max")
    # 4 trials -> all land in DIFFERENT brackets; add 4 more so bracket 0
    # (grace 1) holds t0, t4, t8, t12 as peers
    tids = [f"t{i}" for i in range(16)]
    for tid in tids:
        assert s.on_trial_result(tid, {"score": 0.0,
                                       "training_iteration": 0}) == CONTINUE
    # good peers of bracket 0 hit rung 1 first with high scores
    for tid, v in [("t4", 10.0), ("t8", 11.0), ("t12", 12.0)]:
        assert s.on_trial_result(tid, {"score": v,
                                       "training_iteration": 1}) == CONTINUE
    # the straggler with a terrible score is cut at the rung
    assert s.on_trial_result("t0", {"score": 0.1,
                                    "training_iteration": 1}) == STOP
    # a late GOOD trial in the same bracket survives... (new peer set)
    assert s.on_trial_result("t4", {"score": 99.0,
                                    "training_iteration": 3}) == CONTINUE
    # and everything stops at max_t
    assert s.on_trial_result("t12", {"score": 50.0,
                                     "training_iteration": 27}) == STOP


def test_pb2_gp_exploit_moves_params(ray_mod):
    from ant_ray_amd.tune.schedulers import PB2, PERTURB

    sched = PB2(metric="score", mode="max", perturbation_interval=1,
                hyperparam_bounds={"lr": (0.001, 0.1)}, seed=0)
    sched.set_objective("score", "max")
    for i, tid in enumerate(["a", "b", "c", "d"]):
        sched.on_trial_start(tid, {"lr": 0.01 * (i + 1)})
        sched.on_checkpoint(tid, f"/tmp/ck_{tid}")
    decisions = {}
    for rep in range(3):
        for i, tid in enumerate(["a", "b", "c", "d"]):
            decisions[tid] = sched.on_trial_result(
                tid, {"score": float(i) + rep * 0.1,
                      "training_iteration": rep + 1})
    assert decisions["a"] == PERTURB  # worst trial exploits
    out = sched.exploit("a")
    assert out is not None
    path, cfg = out
    assert 0.001 <= cfg["lr"] <= 0.1
    assert path.startswith("/tmp/ck_")
