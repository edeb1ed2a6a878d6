// app kernels (KGE/w2v/MF) — filled in as models land
