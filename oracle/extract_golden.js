// Extract golden (sk, pk) vectors from harmony-one/harmony's committed .hmy/*.key files.
// Keyfile format (reference internal/blsgen/lib.go:101-159):
//   hex( nonce(12B) || AES-256-GCM(ct) || tag(16B) ), key = ASCII-hex(MD5(passphrase)),
//   passphrase = "" for the committed localnet keys (reference test/deploy.sh:36-40).
// Plaintext = 64-hex-char herumi little-endian Fr secret key.
// Run IN THE DEV CONTAINER ONLY (node v12 present here; not on the GPU box):
//   node oracle/extract_golden.js /root/reference/.hmy > tests/golden/sk_pk.json
const crypto = require('crypto');
const fs = require('fs');
const path = require('path');
const dir = process.argv[2] || '/root/reference/.hmy';
const key = Buffer.from(crypto.createHash('md5').update('').digest('hex'), 'ascii'); // 32 ASCII bytes
const out = [];
for (const f of fs.readdirSync(dir).sort()) {
  if (!f.endsWith('.key')) continue;
  const raw = Buffer.from(fs.readFileSync(path.join(dir, f), 'utf8').trim(), 'hex');
  const nonce = raw.slice(0, 12), tag = raw.slice(raw.length - 16), ct = raw.slice(12, raw.length - 16);
  const d = crypto.createDecipheriv('aes-256-gcm', key, nonce);
  d.setAuthTag(tag);
  const sk = Buffer.concat([d.update(ct), d.final()]).toString('utf8').trim();
  out.push({ pk: f.replace(/\.key$/, ''), sk });
}
console.log(JSON.stringify(out, null, 1));
