"""Observation/reward function ABCs (reference
``ddls/environments/ddls_observation_function.py:5``,
``ddls_reward_function.py:5``, ``ddls_observation.py:3``)."""
from __future__ import annotations

from abc import ABC, abstractmethod


class DDLSObservationFunction(ABC):
    @abstractmethod
    def reset(self, env):
        ...

    @abstractmethod
    def extract(self, env, done: bool):
        ...


class DDLSRewardFunction(ABC):
    def reset(self, env=None, **kwargs):
        return None

    @abstractmethod
    def extract(self, env, done: bool):
        ...


class DDLSObservation(dict):
    """Observations are plain dicts of numpy arrays in this rebuild."""
