import subprocess
import sys

from setuptools import setup, find_packages
from setuptools.command.build_py import build_py


class BuildWithHip(build_py):
    def run(self):
        subprocess.run([sys.executable, '-c',
                        'from chunkflow_amd.build import build; build()'],
                       check=True)
        super().run()


setup(
    name='chunkflow-amd',
    version='0.1.0',
    description="MI355X-native chunkflow inference hot path",
    packages=find_packages(include=['chunkflow_amd*']),
    python_requires='>=3.8',
    entry_points={
        'console_scripts': ['chunkflow = chunkflow_amd.flow:main'],
    },
    cmdclass={'build_py': BuildWithHip},
)
