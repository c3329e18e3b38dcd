"""In-tree build of the qsa_hip extension for MI355X (gfx950).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands inside quickstart_streaming_agents_amd/ so the gpurun
snapshot carries it (it is git-ignored; history stays source-only).
hipcc cross-compiles gfx950 without a GPU present.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, "quickstart_streaming_agents_amd", "ops", "hip")

sources = [
    os.path.join(HIP_DIR, "ops.cpp"),
    os.path.join(HIP_DIR, "avro_codec.cpp"),
    os.path.join(HIP_DIR, "elementwise.hip"),
    os.path.join(HIP_DIR, "paged_attn.hip"),
    os.path.join(HIP_DIR, "skinny_gemm.hip"),
    os.path.join(HIP_DIR, "skinny_gemm_fp8.hip"),
    os.path.join(HIP_DIR, "hash_join.hip"),
    os.path.join(HIP_DIR, "gemm_fp8_batch.hip"),
    os.path.join(HIP_DIR, "topk_cosine.hip"),
    os.path.join(HIP_DIR, "streaming.hip"),
]

from setuptools import find_packages  # noqa: E402

setup(
    name="quickstart-streaming-agents-amd",
    version="0.1.0",
    description=("MI355X-native streaming-agent engine: Kafka-wire "
                 "ingest, Flink-SQL-subset CREATE AGENT surface, "
                 "hand-written CDNA4 HIP kernels, RCCL over xGMI"),
    packages=find_packages(include=["quickstart_streaming_agents_amd*"]),
    package_data={"quickstart_streaming_agents_amd": [
        "labs/sql/*.sql", "data/*.csv", "data/*.json"]},
    python_requires=">=3.10",
    entry_points={"console_scripts": [
        "qsa=quickstart_streaming_agents_amd.cli:main"]},
    ext_modules=[
        CUDAExtension(
            name="quickstart_streaming_agents_amd.qsa_hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
