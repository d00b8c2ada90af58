"""Mirrors reference test_utils.py:7-48."""
import asyncio

import numpy as np
import pytest

from pytensor_federated_amd.utils import argmin_none_or_func, get_useful_event_loop


class TestArgminNoneOrFunc:
    def test_basic(self):
        assert argmin_none_or_func([3, 1, 2], float) == 1
        assert argmin_none_or_func([3, None, 2], float) == 2
        assert argmin_none_or_func([None, None], float) is None
        assert argmin_none_or_func([], float) is None

    def test_func_applied(self):
        assert argmin_none_or_func([{"l": 5}, {"l": 2}, None], lambda d: d["l"]) == 1


class TestGetUsefulEventLoop:
    def test_no_running_loop(self):
        loop = get_useful_event_loop()
        assert isinstance(loop, asyncio.AbstractEventLoop)
        assert not loop.is_running()

    def test_nested_reentrance(self):
        async def inner():
            loop = get_useful_event_loop()
            # the running loop must be patched for re-entrance
            assert hasattr(loop, "_nest_patched")
            # and must be able to run a nested coroutine to completion

            async def nested():
                return 42

            return loop.run_until_complete(nested())

        loop = get_useful_event_loop()
        assert loop.run_until_complete(inner()) == 42
