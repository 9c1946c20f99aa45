"""Environment registry and construction.

Parity with /root/reference/stoix/utils/make_env.py: an ``ENV_MAKERS``
dispatch over suites (:420-433) and ``make(config) -> (train_env, eval_env)``
(:436-466). Suites implemented offline:

  * ``classic``/``gymnax`` — CartPole/Pendulum/MountainCar(+continuous)/
    Acrobot plus the MinAtar grid games (Freeway, SpaceInvaders, Asterix,
    Breakout-MinAtar);
  * ``brax`` — Ant / Humanoid / HalfCheetah / Hopper-class physics;
  * ``jumanji`` — Snake, Game2048, Connector, Sokoban, RobotWarehouse;
  * ``envpool`` — Breakout + Pong via the native C++ batched CPU pool
    (envs/csrc/envpool_cpu.cpp), LunarLander, and the cartpole alias;
  * ``debug`` — five diagnostic games.

  * ``xland_minigrid`` — goal-conditioned procedural gridworld
    (envs/xland.py; capability-class equivalent of the reference's
    JAX-only xminigrid suite);
  * ``craftax`` — crafting/achievement-chain world (envs/crafting.py;
    capability-class equivalent of the JAX-only craftax suite).

Also implemented in-repo (round 2): ``navix`` (DoorKey/Empty grids),
``popjym`` + ``popgym_arcade`` (POMDPs and memory games), ``kinetix``
(procedural reacher slice), ``mujoco_playground`` (swing-up + aliases),
``jaxarc`` (GridCopy slice), ``gymnasium`` (adapter; needs the gymnasium
package). Unknown scenario names raise a clear error listing what is
available.
"""
from __future__ import annotations

from typing import Callable, Dict, Tuple

import torch

from stoix_amd.envs.ant import Ant
from stoix_amd.envs.classic import Acrobot, CartPole, MountainCar, Pendulum
from stoix_amd.envs.debug import DEBUG_ENVIRONMENTS
from stoix_amd.envs.env import StatefulVecEnv, get_final_step_metrics

__all__ = ["make", "make_factory", "StatefulVecEnv", "get_final_step_metrics", "ENV_REGISTRY"]


def _classic(name: str):
    from stoix_amd.envs.minatar import Asterix, BreakoutMinAtar, Freeway, SpaceInvaders

    table = {
        "CartPole-v1": CartPole,
        "Pendulum-v1": Pendulum,
        "MountainCar-v0": MountainCar,
        "MountainCarContinuous-v0": lambda **kw: MountainCar(continuous=True, **kw),
        "Acrobot-v1": Acrobot,
        # MinAtar-class grid games (the reference's gymnax suite)
        "Freeway-MinAtar": Freeway,
        "freeway": Freeway,
        "SpaceInvaders-MinAtar": SpaceInvaders,
        "space_invaders": SpaceInvaders,
        "Asterix-MinAtar": Asterix,
        "asterix": Asterix,
        "Breakout-MinAtar": BreakoutMinAtar,
    }
    if name not in table:
        raise ValueError(f"unknown classic env '{name}' (have {list(table)})")
    return table[name]


def _brax(name: str):
    from stoix_amd.envs.humanoid import Humanoid
    from stoix_amd.envs.planar import HalfCheetah, Hopper

    table = {"ant": Ant, "humanoid": Humanoid, "halfcheetah": HalfCheetah, "hopper": Hopper}
    if name not in table:
        raise ValueError(f"unknown brax-suite env '{name}' (have {list(table)})")
    return table[name]


def _jumanji(name: str):
    from stoix_amd.envs.connector import Connector
    from stoix_amd.envs.game2048 import Game2048
    from stoix_amd.envs.snake import Snake
    from stoix_amd.envs.rware import RobotWarehouse
    from stoix_amd.envs.sokoban import Sokoban

    table = {
        "snake": Snake,
        "Snake-v1": Snake,
        "game_2048": Game2048,
        "2048": Game2048,
        "Game2048-v1": Game2048,
        "connector": Connector,
        "Connector-v2": Connector,
        "sokoban": Sokoban,
        "Sokoban-v0": Sokoban,
        "robot_warehouse": RobotWarehouse,
        "RobotWarehouse-v0": RobotWarehouse,
        "rware": RobotWarehouse,
    }
    if name not in table:
        raise ValueError(f"unknown jumanji-suite env '{name}' (have {list(table)})")
    return table[name]


def _envpool(name: str):
    from stoix_amd.envs.breakout import Breakout

    def breakout_maker(**kw):
        # envpool parity: prefer the native C++ batched CPU engine for
        # CPU actor threads (Sebulba); torch-ops path on CUDA / unbuilt ext
        if torch.device(kw.get("device", "cpu")).type == "cpu":
            from stoix_amd.envs.envpool_cpu import BreakoutCpu, envpool_ext

            if envpool_ext() is not None:
                return BreakoutCpu(**kw)
        return Breakout(**kw)

    from stoix_amd.envs.lunarlander import LunarLander

    def pong_maker(**kw):
        from stoix_amd.envs.envpool_cpu import PongCpu, envpool_ext

        if envpool_ext(required=True) is not None:
            return PongCpu(**kw)

    def spaceinv_maker(**kw):
        from stoix_amd.envs.envpool_cpu import SpaceInvadersCpu, envpool_ext

        if envpool_ext(required=True) is not None:
            return SpaceInvadersCpu(**kw)

    def qbert_maker(**kw):
        from stoix_amd.envs.envpool_cpu import QbertCpu, envpool_ext

        if envpool_ext(required=True) is not None:
            return QbertCpu(**kw)

    def vizdoom_maker(**kw):
        from stoix_amd.envs.envpool_cpu import VizdoomBasicCpu, envpool_ext

        if envpool_ext(required=True) is not None:
            return VizdoomBasicCpu(**kw)

    def phoenix_maker(**kw):
        from stoix_amd.envs.envpool_cpu import PhoenixCpu, envpool_ext

        if envpool_ext(required=True) is not None:
            return PhoenixCpu(**kw)

    def battlezone_maker(**kw):
        from stoix_amd.envs.envpool_cpu import BattlezoneCpu, envpool_ext

        if envpool_ext(required=True) is not None:
            return BattlezoneCpu(**kw)

    def doubledunk_maker(**kw):
        from stoix_amd.envs.envpool_cpu import DoubledunkCpu, envpool_ext

        if envpool_ext(required=True) is not None:
            return DoubledunkCpu(**kw)

    def namethisgame_maker(**kw):
        from stoix_amd.envs.envpool_cpu import NameThisGameCpu, envpool_ext

        if envpool_ext(required=True) is not None:
            return NameThisGameCpu(**kw)

    table = {
        "breakout": breakout_maker,
        "vizdoom_basic": vizdoom_maker,
        "VizdoomBasic-v1": vizdoom_maker,
        "Breakout-v5": breakout_maker,
        "pong": pong_maker,
        "Pong-v5": pong_maker,
        "space_invaders": spaceinv_maker,
        "SpaceInvaders-v5": spaceinv_maker,
        "qbert": qbert_maker,
        "Qbert-v5": qbert_maker,
        "phoenix": phoenix_maker,
        "Phoenix-v5": phoenix_maker,
        "battlezone": battlezone_maker,
        "BattleZone-v5": battlezone_maker,
        "doubledunk": doubledunk_maker,
        "DoubleDunk-v5": doubledunk_maker,
        "namethisgame": namethisgame_maker,
        "NameThisGame-v5": namethisgame_maker,
        # classic-control names envpool also serves
        "cartpole": CartPole,
        "CartPole-v1": CartPole,
        "lunarlander": LunarLander,
        "LunarLander-v2": LunarLander,
    }
    if name not in table:
        raise ValueError(f"unknown envpool-suite env '{name}' (have {list(table)})")
    return table[name]


def _debug(name: str):
    if name not in DEBUG_ENVIRONMENTS:
        raise ValueError(f"unknown debug env '{name}' (have {list(DEBUG_ENVIRONMENTS)})")
    return DEBUG_ENVIRONMENTS[name]


def _xland(name: str):
    from stoix_amd.envs.xland import (
        DoorKeyGrid,
        DoorKeyGrid5,
        DoorKeyGrid8,
        EmptyGrid5,
        EmptyGrid6,
        XLandGrid,
    )

    table = {
        "goal_grid": XLandGrid,
        "XLand-MiniGrid-R1": XLandGrid,
        # navix/MiniGrid-class tasks share the gridworld machinery; the
        # reference's sized scenarios (xland_minigrid/empty_*.yaml,
        # door_key_5x5.yaml; navix/empty_5x5.yaml, door_key_8x8.yaml)
        "doorkey": DoorKeyGrid,
        "Navix-DoorKey-8x8-v0": DoorKeyGrid,
        "empty_5x5": EmptyGrid5,
        "empty_6x6": EmptyGrid6,
        "door_key_5x5": DoorKeyGrid5,
        "door_key_8x8": DoorKeyGrid8,
    }
    if name not in table:
        raise ValueError(f"unknown xland-suite env '{name}' (have {list(table)})")
    return table[name]


def _craftax(name: str):
    from stoix_amd.envs.crafting import Crafting, CraftingPixels

    table = {
        "crafting": Crafting,
        "Craftax-Symbolic-v1": Crafting,
        # the reference's four craftax scenario yamls (symbolic /
        # classic_symbolic / pixels / classic_pixels)
        "symbolic": Crafting,
        "classic_symbolic": Crafting,
        "crafting_pixels": CraftingPixels,
        "pixels": CraftingPixels,
        "classic_pixels": CraftingPixels,
    }
    if name not in table:
        raise ValueError(f"unknown craftax-suite env '{name}' (have {list(table)})")
    return table[name]


def _popjym(name: str):
    from stoix_amd.envs.pomdp import (
        AutoEncodeEasy,
        AutoEncodeMedium,
        CountRecallEasy,
        CountRecallMedium,
        NoisyStatelessCartPole,
        RepeatFirstEasy,
        RepeatFirstHard,
        RepeatFirstMedium,
        StatelessCartPole,
        StatelessPendulum,
    )

    table = {
        "stateless_cartpole": StatelessCartPole,
        "StatelessCartPole": StatelessCartPole,
        # the reference's popjym difficulty-suffixed scenario names
        # (configs/env/popjym/*.yaml: StatelessCartPoleEasy, AutoencodeEasy,
        # CountRecallEasy/Medium, RepeatFirstEasy/Medium/Hard)
        "stateless_cartpole_easy": StatelessCartPole,
        "StatelessCartPoleEasy": StatelessCartPole,
        "noisy_stateless_cartpole": NoisyStatelessCartPole,
        "NoisyStatelessCartPole": NoisyStatelessCartPole,
        "stateless_pendulum": StatelessPendulum,
        "StatelessPendulum": StatelessPendulum,
        "auto_encode_easy": AutoEncodeEasy,
        "AutoencodeEasy": AutoEncodeEasy,
        "auto_encode_medium": AutoEncodeMedium,
        "AutoencodeMedium": AutoEncodeMedium,
        "count_recall_easy": CountRecallEasy,
        "CountRecallEasy": CountRecallEasy,
        "count_recall_medium": CountRecallMedium,
        "CountRecallMedium": CountRecallMedium,
        "repeat_first_easy": RepeatFirstEasy,
        "RepeatFirstEasy": RepeatFirstEasy,
        "repeat_first_medium": RepeatFirstMedium,
        "RepeatFirstMedium": RepeatFirstMedium,
        "repeat_first_hard": RepeatFirstHard,
        "RepeatFirstHard": RepeatFirstHard,
    }
    if name not in table:
        raise ValueError(f"unknown popjym-suite env '{name}' (have {list(table)})")
    return table[name]


def _popgym_arcade(name: str):
    # popgym_arcade suite (reference configs/env/popgym_arcade/
    # noisy_cartpole.yaml: NoisyCartPoleEasy with partial_obs=True) — the
    # partially-observable noisy cartpole maps to the same capability class.
    from stoix_amd.envs.pomdp import NoisyStatelessCartPole

    table = {
        "noisy_cartpole": NoisyStatelessCartPole,
        "NoisyCartPoleEasy": NoisyStatelessCartPole,
    }
    if name not in table:
        raise ValueError(
            f"unknown popgym_arcade-suite env '{name}' (have {list(table)})"
        )
    return table[name]


def _playground(name: str):
    from stoix_amd.envs.classic import CartPoleBalance, CartPoleSwingUp, Pendulum
    from stoix_amd.envs.humanoid import Humanoid
    from stoix_amd.envs.planar import HalfCheetah, Hopper

    # mujoco_playground-class: dm_control-style control + locomotion
    table = {
        "cartpole_swingup": CartPoleSwingUp,
        "CartpoleSwingup": CartPoleSwingUp,
        # dm_control-tier scenarios (reference mjc_playground/dm_control/*)
        "cartpole_balance": CartPoleBalance,
        "CartpoleBalance": CartPoleBalance,
        "hopper_hop": Hopper,
        "HopperHop": Hopper,
        # quadruped joystick locomotion maps to the Ant-class physics tier
        # (reference mjc_playground/locomotion/go_1_joystick_flat_terrain)
        "go_1_joystick_flat_terrain": Ant,
        "Go1JoystickFlatTerrain": Ant,
        "pendulum_swingup": Pendulum,
        # locomotion tasks map onto the physics envs of the brax-class tier
        "ant": Ant,
        "humanoid": Humanoid,
        "halfcheetah": HalfCheetah,
        "hopper": Hopper,
    }
    if name not in table:
        raise ValueError(f"unknown playground-suite env '{name}' (have {list(table)})")
    return table[name]


def _kinetix(name: str):
    from stoix_amd.envs.reacher import (
        ProceduralReacher,
        ProceduralReacher3,
        ProceduralReacherSmall,
    )

    table = {
        "reacher": ProceduralReacher,
        "Kinetix-Reacher-v1": ProceduralReacher,
        # the reference's env_size tiers (configs/env/kinetix/{small,medium,
        # large,all}.yaml) scale scene complexity; here they scale the
        # articulation (link count + morphology range) — documented slice
        "small": ProceduralReacherSmall,
        "s": ProceduralReacherSmall,
        "medium": ProceduralReacher,
        "m": ProceduralReacher,
        "large": ProceduralReacher3,
        "l": ProceduralReacher3,
        "all": ProceduralReacher3,
    }
    if name not in table:
        raise ValueError(f"unknown kinetix-suite env '{name}' (have {list(table)})")
    return table[name]


def _jaxarc(name: str):
    from stoix_amd.envs.arc import GridCopy

    from stoix_amd.envs.arc import GridMirror

    table = {
        "grid_copy": GridCopy,
        "Arc-GridCopy-v1": GridCopy,
        # concept-class transformation slice (reference jaxarc concept
        # grouping): paint the MIRROR of the shown sprite
        "grid_mirror": GridMirror,
        "Arc-GridMirror-v1": GridMirror,
    }
    if name not in table:
        raise ValueError(f"unknown jaxarc-suite env '{name}' (have {list(table)})")
    return table[name]


def _gymnasium(name: str):
    # the gymnasium suite is factory-only (stateful CPU envs for Sebulba,
    # reference utils/env_factory.py:71-86); Anakin's make_single cannot
    # build it without the gymnasium package installed
    def maker(**kw):
        raise ImportError(
            "env suite 'gymnasium' is a Sebulba factory path "
            "(environments.make_factory) and needs the gymnasium package, "
            "which is not installed in this offline image"
        )

    return maker


ENV_REGISTRY: Dict[str, Callable] = {
    "classic": _classic,
    "gymnax": _classic,  # alias: the reference's gymnax suite maps to classic control here
    "brax": _brax,
    "jumanji": _jumanji,
    "envpool": _envpool,
    "gymnasium": _gymnasium,
    "xland_minigrid": _xland,
    "xland": _xland,
    "navix": _xland,
    "mujoco_playground": _playground,
    "kinetix": _kinetix,
    "jaxarc": _jaxarc,
    "craftax": _craftax,
    "popjym": _popjym,
    "popgym_arcade": _popgym_arcade,
    "debug": _debug,
}


def make_single(
    config,
    num_envs: int,
    device: torch.device | str = "cpu",
    seed: int = 0,
) -> StatefulVecEnv:
    env_cfg = config.env
    suite = env_cfg.env_name
    if suite not in ENV_REGISTRY:
        raise ValueError(
            f"env suite '{suite}' is not available in this offline build "
            f"(have {list(ENV_REGISTRY)})"
        )
    scenario = env_cfg.scenario.name if hasattr(env_cfg, "scenario") else env_cfg.scenario_name
    cls = ENV_REGISTRY[suite](scenario)
    kwargs = dict(getattr(env_cfg, "kwargs", {}) or {})
    env = cls(num_envs=num_envs, device=device, seed=seed, **kwargs)
    thr = getattr(env_cfg, "solved_return_threshold", None)
    if thr is not None:
        env.solved_return_threshold = float(thr)
    return env


def make(config, device: torch.device | str = "cpu") -> Tuple[StatefulVecEnv, StatefulVecEnv]:
    """Build (train_env, eval_env) as the reference's ``environments.make``
    does (make_env.py:436-466)."""
    num_envs = int(config.arch.num_envs)
    num_eval = int(getattr(config.arch, "num_eval_episodes", 128))
    seed = int(getattr(config.arch, "seed", 0))
    train_env = make_single(config, num_envs, device, seed)
    eval_env = make_single(config, num_eval, device, seed + 10_000)
    return train_env, eval_env


def make_factory(config, device: torch.device | str = "cpu"):
    """Sebulba env factory: thread-safe builder of fresh CPU vec-envs with
    unique seeds (reference utils/env_factory.py:23-86). The gymnasium
    suite routes to GymnasiumFactory (reference env_factory.py:71-86)."""
    import itertools
    import threading

    base_seed = int(getattr(config.arch, "seed", 0))
    if config.env.env_name == "gymnasium":
        from stoix_amd.envs.gymnasium_adapter import GymnasiumFactory

        env_cfg = config.env
        scenario = env_cfg.scenario.name if hasattr(env_cfg, "scenario") else env_cfg.scenario_name
        kwargs = dict(getattr(env_cfg, "kwargs", {}) or {})
        return GymnasiumFactory(scenario, seed=base_seed, **kwargs)

    counter = itertools.count()
    lock = threading.Lock()

    class _Factory:
        def __call__(self, num_envs: int) -> StatefulVecEnv:
            with lock:
                idx = next(counter)
            return make_single(config, num_envs, device, base_seed + 7919 * (idx + 1))

    return _Factory()
