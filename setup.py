from setuptools import setup, find_packages

setup(
    name='arrow-matrix-amd',
    version='0.1.0',
    description='MI355X-native arrow-decomposition iterated-SpMM engine '
                '(rebuild of spcl/arrow-matrix hot path)',
    packages=find_packages(include=['arrow_matrix_amd', 'arrow_matrix_amd.*',
                                    'scripts']),
    python_requires='>=3.10',
    entry_points={
        'console_scripts': [
            # same entry point names as the reference (setup.py:17-24)
            'arrow_decompose = scripts.decomposition_main:main',
            'spmm_arrow = scripts.spmm_arrow_main:main',
            'spmm_15d = scripts.spmm_15d_main:main',
            'spmm_petsc = scripts.spmm_petsc_main:main',
        ],
    },
)
