"""Agent pipeline resolution (reference `distar/agent/import_helper.py:11-14`
+ README 'pipeline' contract): agents are loaded by string name so
third-party agent packages drop in without touching the framework.

Resolution order for `import_pipeline_module(pipeline, name)`:
  1. `distar_amd.agents.<pipeline>` (in-tree pipelines; 'default' maps to
     the built-in actor/learner classes),
  2. an importable top-level package named `<pipeline>` exposing the same
     class names (external agents, `docs/agent.md` contract).
"""
import importlib

_BUILTIN = {
    'default': {
        'Agent': 'distar_amd.actor.agent',
        'Actor': 'distar_amd.actor.actor',
        'SLLearner': 'distar_amd.learner.sl_learner',
        'RLLearner': 'distar_amd.learner.rl_learner',
        'Model': 'distar_amd.models.alphastar.model',
        'ReplayDecoder': 'distar_amd.data.replay_decoder',
    },
    'bot': {},       # built-in game bots need no python classes
}


def import_pipeline_module(pipeline, name):
    """-> class ``name`` for agent ``pipeline``."""
    if pipeline in _BUILTIN:
        modpath = _BUILTIN[pipeline].get(name)
        if modpath is None:
            raise ImportError(f'pipeline {pipeline!r} has no component {name!r}')
        return getattr(importlib.import_module(modpath), name)
    try:
        mod = importlib.import_module(f'distar_amd.agents.{pipeline}')
    except ImportError:
        mod = importlib.import_module(pipeline)
    return getattr(mod, name)


def import_pipeline_agent(pipeline):
    return import_pipeline_module(pipeline, 'Agent')
