"""Build: in-tree HIP extension (gfx950) + plugin entry points.

The extension is built IN-TREE (`python setup.py build_ext --inplace`) so
the .so travels with repo snapshots to GPU boxes.  hipcc cross-compiles
gfx950 without a GPU present.
"""
import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join("gymfx_amd", "ops", "csrc")

ext = CUDAExtension(
    name="gymfx_amd.ops._gymfx_hip",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "env_step.hip"),
        os.path.join(CSRC, "ppo_kernels.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="gymfx-amd",
    version="0.1.0",
    description=(
        "MI355X-native vectorized FX-trading RL framework "
        "(gym-fx capabilities, HIP/CDNA4 kernels, RCCL data parallelism)"
    ),
    packages=find_packages(include=["gymfx_amd", "gymfx_amd.*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
    python_requires=">=3.9",
    entry_points={
        "console_scripts": [
            "gymfx-amd=gymfx_amd.main:main",
        ],
        "data_feed.plugins": [
            "default_data_feed=gymfx_amd.plugins.data_feeds:DefaultDataFeed",
            "synthetic_data_feed=gymfx_amd.plugins.data_feeds:SyntheticDataFeed",
        ],
        "broker.plugins": [
            "default_broker=gymfx_amd.plugins.brokers:DefaultBroker",
            "oanda_broker=gymfx_amd.plugins.brokers:OandaBroker",
        ],
        "strategy.plugins": [
            "default_strategy=gymfx_amd.plugins.strategies:DefaultStrategy",
            "direct_fixed_sltp=gymfx_amd.plugins.strategies:DirectFixedSLTP",
            "direct_atr_sltp=gymfx_amd.plugins.strategies:DirectAtrSLTP",
        ],
        "preprocessor.plugins": [
            "default_preprocessor=gymfx_amd.plugins.preprocessors:DefaultPreprocessor",
            "feature_window_preprocessor=gymfx_amd.plugins.preprocessors:FeatureWindowPreprocessor",
        ],
        "reward.plugins": [
            "pnl_reward=gymfx_amd.plugins.rewards:PnlReward",
            "sharpe_reward=gymfx_amd.plugins.rewards:SharpeReward",
            "dd_penalized_reward=gymfx_amd.plugins.rewards:DdPenalizedReward",
        ],
        "metrics.plugins": [
            "default_metrics=gymfx_amd.plugins.metrics:DefaultMetrics",
            "trading_metrics=gymfx_amd.plugins.metrics:TradingMetrics",
        ],
    },
)
