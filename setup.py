"""Build script: package + the optional native hot-path extension.

`python setup.py build_ext --inplace` compiles native/_amcore.c into
active_monitor_amd/ so the in-tree .so ships with source checkouts. The
package works without it (pure-Python fallback in utils/fastcopy.py)."""
import os

from setuptools import Extension, setup
from setuptools.command.build_ext import build_ext


class OptionalBuildExt(build_ext):
    """Compile the extension when a toolchain exists; never fail the install."""

    def run(self):
        try:
            super().run()
        except Exception as e:  # pragma: no cover
            print(f"warning: skipping native extension build: {e}")

    def build_extension(self, ext):
        try:
            super().build_extension(ext)
        except Exception as e:  # pragma: no cover
            print(f"warning: skipping {ext.name}: {e}")


setup(
    ext_modules=[
        Extension(
            "active_monitor_amd._amcore",
            sources=[os.path.join("native", "_amcore.c")],
            extra_compile_args=["-O2"],
        )
    ],
    cmdclass={"build_ext": OptionalBuildExt},
)
