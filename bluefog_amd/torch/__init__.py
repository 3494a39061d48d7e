# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Alias package so reference-style ``import bluefog_amd.torch as bf``
works verbatim (the reference spelled it ``bluefog.torch``)."""
from bluefog_amd import *  # noqa: F401,F403
from bluefog_amd import init, shutdown  # noqa: F401
