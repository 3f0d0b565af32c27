"""Self-checking GPU workloads for nvshare-amd tests and benchmarks.

MI355X-native ports of the reference test workloads
(reference tests/pytorch-add.py, tests/tf-matmul.py and their -small
variants), plus a ResNet-50 training job and a bursty inference job for
the co-location scenarios BASELINE.json names.  Every workload prints
"PASS <seconds>" on success and a final JSON line with its metrics.
"""
