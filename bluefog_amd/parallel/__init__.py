# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Distribution strategies: virtual topologies (static graph families +
dynamic one-peer generators) and topology inference."""
