"""Hot columnar operators: HIP/CDNA4 kernels with torch CPU reference
backends. See SURVEY.md §2.10 for the kernel inventory K1-K21."""
