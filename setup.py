from setuptools import find_packages, setup

setup(
    name="kukeon-amd",
    version="0.1.0",
    description="MI355X-native self-hosted AI-agent runtime "
                "(kukeon capabilities, gfx950 data plane)",
    packages=find_packages(include=["kukeon_amd*"]),
    python_requires=">=3.10",
    entry_points={
        "console_scripts": [
            "kuke=kukeon_amd.cli.main:main",
            "kukeond=kukeon_amd.cli.main:main",
        ]
    },
)
