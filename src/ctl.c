/*
 * nvsharectl — configuration CLI for the nvshare-amd scheduler.
 *
 * Behavioral parity reference: /root/reference/src/cli.c (-T set TQ,
 * -S on|off). Fresh implementation on getopt_long (no vendored parser).
 * nvshare-amd addition: -q queries live scheduler status.
 */
#define _GNU_SOURCE
#include <getopt.h>
#include <inttypes.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <unistd.h>

#include "common.h"
#include "proto.h"

static void usage(const char *argv0)
{
	fprintf(stderr,
		"Usage: %s [OPTION]...\n"
		"Configure the nvshare-amd scheduler.\n\n"
		"  -T, --set-tq SECONDS    set the time quantum (>= 1)\n"
		"  -g, --gpu N             apply -T to one GPU only "
		"(put -g before -T)\n"
		"  -S, --scheduler on|off  enable/disable anti-thrashing "
		"scheduling\n"
		"  -q, --status            print scheduler status\n"
		"  -h, --help              show this help\n\n"
		"The scheduler socket is <NVSHARE_SOCK_DIR>/scheduler.sock "
		"(default " NVS_SOCK_DIR_DEFAULT ").\n",
		argv0);
}

static int send_one(uint8_t type, const char *data, int want_reply)
{
	char path[NVS_SOCK_PATH_MAX];
	struct nvs_msg m;
	int fd;

	if (nvs_scheduler_path(path) != 0) {
		fprintf(stderr, "nvsharectl: socket path too long\n");
		return 1;
	}
	fd = nvs_connect_path(path);
	if (fd < 0) {
		fprintf(stderr, "nvsharectl: cannot connect to %s "
			"(is nvshare-scheduler running?)\n", path);
		return 1;
	}
	nvs_msg_init(&m, type, 0, data);
	if (nvs_send_msg(fd, &m) != 0) {
		fprintf(stderr, "nvsharectl: send failed\n");
		close(fd);
		return 1;
	}
	if (want_reply) {
		if (nvs_recv_msg(fd, &m) != 0 || m.type != NVS_STATUS) {
			fprintf(stderr, "nvsharectl: no status reply\n");
			close(fd);
			return 1;
		}
		{
			int on = 0, tq = 0, ncl = 0, qlen = 0;
			long mib = 0;
			char buf[NVS_MSG_DATA_LEN];
			int nf;

			memcpy(buf, m.data, NVS_MSG_DATA_LEN);
			buf[NVS_MSG_DATA_LEN - 1] = '\0';
			nf = sscanf(buf, "%d,%d,%d,%d,%ld", &on, &tq,
				    &ncl, &qlen, &mib);
			if (nf >= 4) {
				printf("scheduling: %s\ntq: %d s\n"
				       "clients: %d\nqueued: %d\n",
				       on ? "on" : "off", tq, ncl, qlen);
				if (nf >= 5)
					printf("tracked memory: %ld MiB\n",
					       mib);
			} else {
				printf("status: %s\n", buf);
			}
		}
	}
	close(fd);
	return 0;
}

int main(int argc, char **argv)
{
	static const struct option longopts[] = {
		{ "set-tq", required_argument, NULL, 'T' },
		{ "scheduler", required_argument, NULL, 'S' },
		{ "gpu", required_argument, NULL, 'g' },
		{ "status", no_argument, NULL, 'q' },
		{ "help", no_argument, NULL, 'h' },
		{ NULL, 0, NULL, 0 },
	};
	int opt, did_something = 0, rc = 0;
	long gpu = -1;

	nvs_log_init();
	while ((opt = getopt_long(argc, argv, "T:S:g:qh", longopts, NULL))
	       != -1) {
		switch (opt) {
		case 'g':
			gpu = strtol(optarg, NULL, 10);
			break;
		case 'T': {
			char *end = NULL;
			long v = strtol(optarg, &end, 10);
			char data[32];

			if (end == optarg || *end != '\0' || v < 1 ||
			    v > 86400) {
				fprintf(stderr, "nvsharectl: invalid TQ "
					"'%s' (want 1..86400 seconds)\n",
					optarg);
				return 1;
			}
			if (gpu >= 0)
				snprintf(data, sizeof(data), "gpu%ld:%ld",
					 gpu, v);
			else
				snprintf(data, sizeof(data), "%ld", v);
			rc |= send_one(NVS_SET_TQ, data, 0);
			did_something = 1;
			break;
		}
		case 'S':
			if (strcmp(optarg, "on") == 0)
				rc |= send_one(NVS_SCHED_ON, NULL, 0);
			else if (strcmp(optarg, "off") == 0)
				rc |= send_one(NVS_SCHED_OFF, NULL, 0);
			else {
				fprintf(stderr, "nvsharectl: -S takes "
					"'on' or 'off'\n");
				return 1;
			}
			did_something = 1;
			break;
		case 'q':
			rc |= send_one(NVS_STATUS_REQ, NULL, 1);
			did_something = 1;
			break;
		case 'h':
			usage(argv[0]);
			return 0;
		default:
			usage(argv[0]);
			return 1;
		}
	}
	if (!did_something) {
		usage(argv[0]);
		return 1;
	}
	return rc;
}
