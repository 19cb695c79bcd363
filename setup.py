"""Editable/sdist install; the gfx950 extension is built in-tree with
``python -m raft_amd.ops.build`` (hipcc --offload-arch=gfx950) and ships
as package data. See pyproject.toml for metadata."""
from setuptools import setup

setup()
