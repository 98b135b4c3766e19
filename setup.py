import os

from setuptools import find_packages, setup

try:
    from torch.utils.cpp_extension import BuildExtension  # noqa: F401
    HAVE_TORCH = True
except ImportError:
    HAVE_TORCH = False


class BuildHipInTree:
    """`python setup.py build_ext --inplace` -> drive hipcc directly."""


def _build_hip():
    os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')
    from coinstac_dinunet_amd.ops.build import build
    build(verbose=True)


if __name__ == '__main__':
    import sys
    if 'build_ext' in sys.argv and HAVE_TORCH:
        _build_hip()
        sys.argv = [a for a in sys.argv if a not in ('build_ext', '--inplace')]
        if len(sys.argv) == 1:
            sys.exit(0)
    setup(
        name='coinstac_dinunet_amd',
        version='0.1.0',
        description=('MI355X-native decentralized federated training with '
                     'the capabilities of coinstac-dinunet'),
        packages=find_packages(include=['coinstac_dinunet_amd*']),
        package_data={'coinstac_dinunet_amd.ops': ['*.so', 'csrc/*']},
        python_requires='>=3.8',
        # torch (ROCm build) is expected from the environment, like the
        # reference (setup.py:34 lists only pure-python deps)
        install_requires=['numpy'],
        extras_require={
            'vision': ['pillow', 'matplotlib', 'scipy'],
            'metrics': ['scikit-learn'],
        },
    )
