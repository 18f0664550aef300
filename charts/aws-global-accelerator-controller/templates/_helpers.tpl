{{- define "agac.name" -}}
{{ .Chart.Name }}
{{- end }}
