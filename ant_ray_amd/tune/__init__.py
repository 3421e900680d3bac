"""ant_ray_amd.tune — Ray Tune parity: hyperparameter search over trials.

Role parity: reference python/ray/tune/ (~58k LoC): Tuner
(tune/tune.py), trial execution (execution/tune_controller.py) on actors,
search (search/), schedulers (schedulers/), ResultGrid. Trials run as
actors (one per trial, resources via with_resources); tune.report shares
the Train session so a function trainable works under both.
"""
from __future__ import annotations

import os
import time
import traceback
from dataclasses import dataclass
from typing import Any, Callable, Dict, List, Optional

from ant_ray_amd.train._checkpoint import Checkpoint
from ant_ray_amd.train.config import Result, RunConfig
from ant_ray_amd.train.session import get_checkpoint  # noqa: F401
from ant_ray_amd.train.session import report  # noqa: F401 (tune.report parity)
from ant_ray_amd.tune.schedulers import (  # noqa: F401
    CONTINUE,
    STOP,
    ASHAScheduler,
    FIFOScheduler,
    HyperBandForBOHB,
    HyperBandScheduler,
    MedianStoppingRule,
    PB2,
    PopulationBasedTraining,
    ResourceChangingScheduler,
)
from ant_ray_amd.tune.search import (  # noqa: F401
    BasicVariantGenerator,
    choice,
    grid_search,
    lograndint,
    loguniform,
    qlograndint,
    qloguniform,
    qrandint,
    qrandn,
    quniform,
    randint,
    randn,
    sample_from,
    uniform,
)
from ant_ray_amd.tune.analysis import ExperimentAnalysis  # noqa: F401
from ant_ray_amd.tune.progress_reporter import (  # noqa: F401
    CLIReporter,
    JupyterNotebookReporter,
    ProgressReporter,
)
from ant_ray_amd.tune.registry import (  # noqa: F401
    create_scheduler,
    create_searcher,
    register_env,
    register_trainable,
)
from ant_ray_amd.tune.stopper import Stopper  # noqa: F401
from ant_ray_amd.tune.trainable import Trainable  # noqa: F401
from ant_ray_amd.train.config import (  # noqa: F401 (shared AIR configs)
    CheckpointConfig,
    FailureConfig,
    SyncConfig,
)


class TuneError(Exception):
    """Parity: reference tune/error.py TuneError."""


@dataclass
class ResumeConfig:
    """Parity: tune/tuner.py ResumeConfig — what to do with unfinished /
    errored trials on Tuner.restore."""

    resume_unfinished: bool = True
    resume_errored: bool = False
    restart_errored: bool = False


class PlacementGroupFactory:
    """Parity: tune/execution/placement_groups.py — bundle list a
    trainable's trial reserves. with_resources() accepts one; the first
    bundle is the trial actor's own resources."""

    def __init__(self, bundles, strategy: str = "PACK"):
        if not bundles:
            raise ValueError("PlacementGroupFactory needs >=1 bundle")
        self.bundles = [dict(b) for b in bundles]
        self.strategy = strategy

    @property
    def head_bundle(self):
        return self.bundles[0]


class TuneContext:
    """Parity: tune/context.py TuneContext (tune.get_context()) — trial
    info inside a trainable function."""

    def __init__(self, train_ctx):
        self._ctx = train_ctx

    def get_trial_id(self) -> str:
        return self._ctx.experiment_name

    def get_trial_name(self) -> str:
        return self._ctx.experiment_name

    def get_trial_dir(self) -> str:
        return self._ctx.experiment_path

    def get_experiment_name(self) -> str:
        return os.path.basename(
            os.path.dirname(self._ctx.experiment_path.rstrip("/")))


def get_context() -> TuneContext:
    from ant_ray_amd.train.session import get_context as _train_get

    return TuneContext(_train_get())


__all__ = [
    "ASHAScheduler", "BasicVariantGenerator", "CLIReporter", "Callback",
    "CheckpointConfig", "Experiment", "ExperimentAnalysis", "FIFOScheduler",
    "FailureConfig", "HyperBandForBOHB", "HyperBandScheduler",
    "JupyterNotebookReporter", "MedianStoppingRule", "PB2",
    "PlacementGroupFactory", "ProgressReporter",
    "ResourceChangingScheduler", "ResultGrid", "ResumeConfig",
    "Stopper", "SyncConfig", "Trainable", "TuneConfig", "TuneContext",
    "TuneError", "Tuner", "choice", "create_scheduler", "create_searcher",
    "get_context", "grid_search", "lograndint", "loguniform", "qlograndint",
    "qloguniform", "qrandint", "qrandn", "quniform", "randint", "randn",
    "register_env", "register_trainable", "report", "run",
    "run_experiments", "sample_from", "uniform", "with_parameters",
    "with_resources",
]


class Callback:
    """Parity: tune/callback.py Callback — hooks fired by the Tuner loop.
    Override any of the on_* methods."""

    def setup(self, **info):
        pass

    def on_trial_start(self, iteration, trials, trial, **info):
        pass

    def on_trial_result(self, iteration, trials, trial, result, **info):
        pass

    def on_trial_complete(self, iteration, trials, trial, **info):
        pass

    def on_trial_error(self, iteration, trials, trial, **info):
        pass

    def on_checkpoint(self, iteration, trials, trial, checkpoint, **info):
        pass

    def on_experiment_end(self, trials, **info):
        pass


@dataclass
class TuneConfig:
    metric: Optional[str] = None
    mode: Optional[str] = "max"
    num_samples: int = 1
    max_concurrent_trials: Optional[int] = None
    scheduler: Optional[Any] = None
    search_alg: Optional[Any] = None
    seed: Optional[int] = None


class ResultGrid:
    def __init__(self, results: List[Result]):
        self._results = results

    def __len__(self):
        return len(self._results)

    def __getitem__(self, i) -> Result:
        return self._results[i]

    def __iter__(self):
        return iter(self._results)

    @property
    def errors(self):
        return [r.error for r in self._results if r.error]

    def get_best_result(self, metric: Optional[str] = None,
                        mode: str = "max") -> Result:
        scored = [r for r in self._results
                  if r.metrics and r.metrics.get(metric) is not None]
        if not scored:
            raise RuntimeError(f"no trial reported metric {metric!r}")
        return (max if mode == "max" else min)(
            scored, key=lambda r: r.metrics[metric])

    def get_dataframe(self):
        import pandas as pd

        return pd.DataFrame([r.metrics or {} for r in self._results])


def with_parameters(fn: Callable, **params):
    """Bind large objects by value (reference puts them in the object store;
    our serializer ships them with the trainable)."""
    import functools

    @functools.wraps(fn)
    def inner(config):
        return fn(config, **params)

    return inner


def with_resources(fn: Callable, resources: Dict[str, float]):
    fn = _copy_fn(fn)
    fn._tune_resources = dict(resources)
    return fn


def _copy_fn(fn):
    import functools

    @functools.wraps(fn)
    def inner(config):
        return fn(config)

    return inner


class _TrialRunner:
    """Actor hosting one trial's function trainable on a thread."""

    def __init__(self):
        import queue
        import threading

        self._q = queue.Queue()
        self._thread = None
        self._error = None
        self._done = False
        self._threading = threading

    def start(self, fn, config, trial_id: str, experiment_path: str,
              restore_path: str = None):
        from ant_ray_amd.train._checkpoint import Checkpoint
        from ant_ray_amd.train.session import TrainContext, set_train_context

        ctx = TrainContext(experiment_name=trial_id,
                           experiment_path=experiment_path)
        ctx.report_queue = self._q
        ctx.stop_requested = self._threading.Event()
        self._stop_ev = ctx.stop_requested
        if restore_path:
            ctx.restore_checkpoint = Checkpoint(restore_path)

        def run():
            set_train_context(ctx)
            try:
                fn(config)
            except SystemExit:
                pass  # cooperative stop via request_stop()
            except BaseException:
                self._error = traceback.format_exc()
            finally:
                self._done = True

        self._thread = self._threading.Thread(target=run, daemon=True)
        self._thread.start()
        return True

    def request_stop(self):
        """Cooperative stop: the next tune.report() in the trial raises
        SystemExit (parity: the reference stops a trial at its next
        result boundary rather than mid-step)."""
        ev = getattr(self, "_stop_ev", None)
        if ev is not None:
            ev.set()
        return True

    def poll(self):
        reports = []
        while True:
            try:
                reports.append(self._q.get_nowait())
            except Exception:
                break
        status = ("errored" if self._error else
                  "finished" if self._done else "running")
        return {"status": status, "reports": reports, "error": self._error}


class Tuner:
    def __init__(self, trainable: Callable, *, param_space: Optional[dict] = None,
                 tune_config: Optional[TuneConfig] = None,
                 run_config: Optional[RunConfig] = None):
        self.trainable = trainable
        self.param_space = param_space or {}
        self.tune_config = tune_config or TuneConfig()
        self.run_config = run_config or RunConfig()
        self._restore_state: Optional[dict] = None

    @classmethod
    def restore(cls, path: str, trainable: Callable,
                resume_errored: bool = False,
                restart_errored: bool = False) -> "Tuner":
        """Resume an interrupted experiment from its directory (parity:
        Tuner.restore — finished trials keep their results, unfinished
        ones re-run from their latest checkpoint; errored trials re-run
        from checkpoint with resume_errored=True or from scratch with
        restart_errored=True)."""
        import pickle

        state_file = os.path.join(path, "tuner_state.pkl")
        with open(state_file, "rb") as f:
            state = pickle.load(f)
        t = cls(trainable, tune_config=state["tune_config"],
                run_config=state["run_config"])
        t.run_config.name = os.path.basename(path.rstrip("/"))
        t.run_config.storage_path = os.path.dirname(path.rstrip("/"))
        state["resume_errored"] = resume_errored
        state["restart_errored"] = restart_errored
        t._restore_state = state
        return t

    def _save_state(self, exp_path, variants, results, running):
        import pickle

        trials = []
        for idx, cfg in enumerate(variants):
            row = {"idx": idx, "config": cfg}
            if idx in results:
                r = results[idx]
                row["status"] = "errored" if r.error else "finished"
                row["metrics"] = r.metrics
                row["ckpt"] = r.checkpoint.path if r.checkpoint else None
            else:
                t = running.get(idx)
                row["status"] = "running" if t else "pending"
                row["ckpt"] = t["ckpt"] if t else None
            trials.append(row)
        tmp = os.path.join(exp_path, ".tuner_state.tmp")
        with open(tmp, "wb") as f:
            pickle.dump({"tune_config": self.tune_config,
                         "run_config": self.run_config,
                         "variants": variants,
                         "trials": trials}, f)
        os.replace(tmp, os.path.join(exp_path, "tuner_state.pkl"))

    def fit(self) -> ResultGrid:
        import ant_ray_amd as ray

        if not ray.is_initialized():
            ray.init()
        if isinstance(self.trainable, str):
            from ant_ray_amd.tune.registry import get_trainable_cls

            self.trainable = get_trainable_cls(self.trainable)
        if isinstance(self.trainable, type) and issubclass(self.trainable,
                                                           Trainable):
            from ant_ray_amd.tune.trainable import _wrap_trainable_cls

            self.trainable = _wrap_trainable_cls(self.trainable,
                                                 stop=self.run_config.stop)
        tc = self.tune_config
        stop_cfg = self.run_config.stop
        callbacks = list(self.run_config.callbacks or [])
        reporter = self.run_config.progress_reporter

        def _stop_hit(trial_id, metrics):
            if stop_cfg is None:
                return False
            if isinstance(stop_cfg, dict):
                return any(metrics.get(k) is not None and metrics[k] >= v
                           for k, v in stop_cfg.items())
            return bool(stop_cfg(trial_id, metrics))

        def _fire_cb(hook, *a, **kw):
            for cb in callbacks:
                try:
                    getattr(cb, hook, lambda *x, **y: None)(*a, **kw)
                except Exception:
                    pass

        def _trial_rows():
            rows = []
            for i, t in running.items():
                rows.append({"trial_id": t["trial_id"], "status": "running",
                             "metrics": t["last_metrics"]})
            for i, r in results.items():
                rows.append({"trial_id": f"{name}_{i:05d}",
                             "status": "errored" if r.error else "finished",
                             "metrics": r.metrics})
            return rows
        search_alg = tc.search_alg
        if isinstance(search_alg, str):
            search_alg = create_searcher(search_alg)
        if self._restore_state is not None:
            variants = self._restore_state["variants"]
            search_alg = None  # restored runs replay the recorded variants
        elif search_alg is not None:
            # sequential suggest/observe searcher (BayesOpt/TPE/wrappers):
            # trials are created on demand, up to num_samples
            search_alg.set_search_properties(tc.metric, tc.mode or "max",
                                             self.param_space)
            variants = []
        else:
            variants = BasicVariantGenerator(
                self.param_space, tc.num_samples, seed=tc.seed).variants()
        scheduler = tc.scheduler or FIFOScheduler()
        scheduler.set_objective(tc.metric, tc.mode or "max")
        name = self.run_config.name or f"tune_{int(time.time())}"
        exp_path = os.path.join(self.run_config.storage_path, name)
        os.makedirs(exp_path, exist_ok=True)
        resources = getattr(self.trainable, "_tune_resources", {"CPU": 1})
        max_conc = tc.max_concurrent_trials or 8

        RunnerCls = ray.remote(_TrialRunner)
        opts = {"num_cpus": resources.get("CPU", 1), "max_concurrency": 2}
        if resources.get("GPU"):
            opts["num_gpus"] = resources["GPU"]

        pending = list(enumerate(variants))
        running: Dict[int, dict] = {}
        results: Dict[int, Result] = {}
        restore_ckpts: Dict[int, str] = {}
        if self._restore_state is not None:
            pending = []
            resume_err = self._restore_state.get("resume_errored")
            restart_err = self._restore_state.get("restart_errored")
            for row in self._restore_state["trials"]:
                idx = row["idx"]
                if row["status"] == "errored" and (resume_err or restart_err):
                    pending.append((idx, row["config"]))
                    if resume_err and row.get("ckpt"):
                        restore_ckpts[idx] = row["ckpt"]
                elif row["status"] in ("finished", "errored"):
                    results[idx] = Result(
                        metrics=row.get("metrics"),
                        checkpoint=Checkpoint(row["ckpt"]) if row.get("ckpt")
                        else None,
                        path=os.path.join(exp_path, f"{name}_{idx:05d}"),
                        error=None if row["status"] == "finished"
                        else RuntimeError("errored before restore"),
                    )
                else:
                    pending.append((idx, row["config"]))
                    if row.get("ckpt"):
                        restore_ckpts[idx] = row["ckpt"]

        def launch(idx, config, restore=None, prev=None):
            restore = restore or restore_ckpts.pop(idx, None)
            trial_id = f"{name}_{idx:05d}"
            trial_path = os.path.join(exp_path, trial_id)
            os.makedirs(trial_path, exist_ok=True)
            actor = RunnerCls.options(**opts).remote()
            ray.get(actor.start.remote(self.trainable, config, trial_id,
                                       trial_path, restore))
            running[idx] = {"actor": actor, "config": config,
                            "trial_id": trial_id, "path": trial_path,
                            "iter": prev["iter"] if prev else 0,
                            "last_metrics": prev["last_metrics"] if prev else None,
                            "ckpt": prev["ckpt"] if prev else None}
            if hasattr(scheduler, "on_trial_start"):
                scheduler.on_trial_start(trial_id, config)
            _fire_cb("on_trial_start", 0, None, trial_id)

        try:
            self._save_state(exp_path, variants, results, running)
        except Exception:
            pass
        suggested = [len(variants)]

        def _more_suggestions():
            return (search_alg is not None
                    and suggested[0] < tc.num_samples)

        while pending or running or _more_suggestions():
            while pending and len(running) < max_conc:
                idx, cfg = pending.pop(0)
                launch(idx, cfg)
            while _more_suggestions() and len(running) < max_conc:
                idx = suggested[0]
                cfg = search_alg.suggest(f"{name}_{idx:05d}")
                if cfg is None:
                    break  # searcher is concurrency-limited right now
                variants.append(cfg)
                suggested[0] += 1
                launch(idx, cfg)
            if not running:
                if pending or _more_suggestions():
                    time.sleep(0.05)
                    continue
                break
            polls = ray.get([t["actor"].poll.remote()
                             for t in running.values()])
            finished_idx = []
            for (idx, t), st in zip(list(running.items()), polls):
                stop = False
                perturb = False
                for rep in st["reports"]:
                    t["iter"] += 1
                    metrics = dict(rep["metrics"])
                    metrics.setdefault("training_iteration", t["iter"])
                    metrics["config"] = t["config"]
                    t["last_metrics"] = metrics
                    if rep.get("checkpoint_path"):
                        t["ckpt"] = rep["checkpoint_path"]
                        if hasattr(scheduler, "on_checkpoint"):
                            scheduler.on_checkpoint(t["trial_id"],
                                                    rep["checkpoint_path"])
                    decision = scheduler.on_trial_result(t["trial_id"], metrics)
                    if decision == STOP:
                        stop = True
                    elif decision == "PERTURB":
                        perturb = True
                    if _stop_hit(t["trial_id"], metrics):
                        stop = True
                    _fire_cb("on_trial_result", t["iter"], None,
                             t["trial_id"], metrics)
                if perturb and st["status"] == "running":
                    # PBT exploit/explore: clone a top trial's checkpoint,
                    # mutate hyperparams, relaunch this slot
                    ex = scheduler.exploit(t["trial_id"])
                    if ex is not None:
                        restore_path, new_cfg = ex
                        try:
                            ray.kill(t["actor"])
                        except Exception:
                            pass
                        launch(idx, new_cfg, restore=restore_path, prev=t)
                        continue
                if stop and st["status"] == "running":
                    try:
                        t["actor"].request_stop.remote()
                    except Exception:
                        pass
                if st["status"] in ("finished", "errored") or stop:
                    err = None
                    if st["status"] == "errored":
                        err = RuntimeError(st["error"])
                    results[idx] = Result(
                        metrics=t["last_metrics"],
                        checkpoint=Checkpoint(t["ckpt"]) if t["ckpt"] else None,
                        path=t["path"], error=err,
                    )
                    _fire_cb("on_trial_error" if err else "on_trial_complete",
                             t["iter"], None, t["trial_id"])
                    if search_alg is not None:
                        try:
                            search_alg.on_trial_complete(
                                t["trial_id"], t["last_metrics"],
                                error=err is not None)
                        except Exception:
                            pass
                    try:
                        ray.kill(t["actor"])
                    except Exception:
                        pass
                    finished_idx.append(idx)
            for idx in finished_idx:
                del running[idx]
            if finished_idx or not running:
                try:
                    self._save_state(exp_path, variants, results, running)
                except Exception:
                    pass
            if reporter is not None and reporter.should_report(
                    _trial_rows(), done=not (pending or running)):
                try:
                    reporter.report(_trial_rows(), not (pending or running))
                except Exception:
                    pass
            if running:
                time.sleep(0.1)
        _fire_cb("on_experiment_end", _trial_rows())
        return ResultGrid([results[i] for i in sorted(results)])


def run(trainable, *, config: Optional[dict] = None, num_samples: int = 1,
        metric: Optional[str] = None, mode: str = "max", scheduler=None,
        **kwargs) -> ResultGrid:
    """Legacy tune.run API (parity tune/tune.py:run)."""
    tuner = Tuner(
        trainable, param_space=config or {},
        tune_config=TuneConfig(metric=metric, mode=mode,
                               num_samples=num_samples, scheduler=scheduler),
        run_config=RunConfig(stop=kwargs.pop("stop", None),
                             name=kwargs.pop("name", None)),
    )
    return tuner.fit()


@dataclass
class Experiment:
    """Parity: tune/experiment/experiment.py — a named spec for
    run_experiments."""

    name: str
    run: Any  # trainable (fn / Trainable subclass / registered name)
    config: Optional[dict] = None
    stop: Optional[Any] = None
    num_samples: int = 1


def run_experiments(experiments, **_) -> Dict[str, ResultGrid]:
    """Run one or more Experiment specs; returns {name: ResultGrid}
    (parity tune/tune.py run_experiments)."""
    if isinstance(experiments, Experiment):
        experiments = [experiments]
    out = {}
    for e in experiments:
        tuner = Tuner(
            e.run, param_space=e.config or {},
            tune_config=TuneConfig(num_samples=e.num_samples),
            run_config=RunConfig(name=e.name, stop=e.stop),
        )
        out[e.name] = tuner.fit()
    return out
