"""Build/install: `python setup.py build_ext --inplace` compiles the
CDNA4 HIP extension in-tree (gfx950) via hipcc."""
from __future__ import annotations

from setuptools import find_packages, setup
from setuptools.command.build_ext import build_ext as _build_ext


class HipBuildExt(_build_ext):
    def run(self):
        from infomesh_amd.ops import _build
        _build.build(force=False)


setup(
    name="infomesh-amd",
    version="0.1.0",
    description="MI355X-native hybrid search/RAG engine",
    packages=find_packages(include=["infomesh_amd", "infomesh_amd.*"]),
    python_requires=">=3.10",
    entry_points={"console_scripts": [
        "infomesh-amd = infomesh_amd.cli:main"]},
    cmdclass={"build_ext": HipBuildExt},
    # the extension is built by hipcc directly; declare a placeholder so
    # `build_ext` runs
    ext_modules=[],
)
