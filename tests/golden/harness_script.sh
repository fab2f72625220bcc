export DEBIAN_FRONTEND=noninteractive
mkdir -p /etc/clawker
printf %s ZWNobyAicG9zdC1pbml0IGhvb2sgZXhlY3V0ZWQiID4+IC90bXAvY2xhd2tlci1ob29rcy5sb2cK | base64 -d > /etc/clawker/post-init.sh
chmod 755 /etc/clawker/post-init.sh

mkdir -p /etc/clawker
printf %s ZWNobyAicHJlLXJ1biBob29rIGV4ZWN1dGVkIiA+PiAvdG1wL2NsYXdrZXItaG9va3MubG9nCg== | base64 -d > /etc/clawker/pre-run.sh
chmod 755 /etc/clawker/pre-run.sh

mkdir -p /etc/clawker
printf %s aGFybmVzczogZWNobwpydWxlczogW10K | base64 -d > /etc/clawker/egress-floor.yaml
chmod 644 /etc/clawker/egress-floor.yaml
