# nvshare-amd top-level build.
#   make            - native components (C) + gfx950 kernels
#   make test       - CPU test suite
#   make tsan       - ThreadSanitizer build of the daemon (race checks)
#   make clean

all:
	$(MAKE) -C src
	$(MAKE) -C hip

test: all
	python3 -m pytest tests -q -m "not gpu"

tsan:
	$(MAKE) -C src BUILD=build-tsan \
	    CFLAGS="-O1 -g -Wall -Wextra -std=gnu11 -fPIC -fsanitize=thread -DNVSHARE_NO_DLSYM_EXPORT" \
	    LDFLAGS_HARDEN="-fsanitize=thread" \
	    build-tsan/nvshare-scheduler build-tsan/libnvshare.so \
	    build-tsan/libamdhip64.so.7 build-tsan/hipclient

clean:
	$(MAKE) -C src clean
	$(MAKE) -C hip clean
	rm -rf src/build-tsan

.PHONY: all test tsan clean
